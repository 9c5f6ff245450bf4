"""On-device validation for the EXPERIMENTAL 256^2 8-phase GEMM
(csrc/gemm_256_8phase.hip). Run on an MI355X box:

    python scripts/validate_gemm256.py            # refcheck + race screen
    python scripts/validate_gemm256.py --bench    # + A/B vs gemm_bt/torch

Guide discipline for NEW sync structures: refcheck at small shapes,
multi-run race screen at 256/512/4096 (sync bugs are timing-dependent —
a single pass can miss them), then A/B within the same process."""

import argparse
import json
import sys
import time

import torch

sys.path.insert(0, ".")
from turboprune_amd.ops._backend import extension  # noqa: E402


def maxerr(got, ref):
    return (got.float() - ref.float()).abs().max().item()


def refcheck(ext, M, N, K, runs=1, seed=0):
    torch.manual_seed(seed)
    A = (torch.rand(M, K, device="cuda") - 0.5).bfloat16()
    B = (torch.rand(N, K, device="cuda") - 0.5).bfloat16()
    ref = (A.float() @ B.float().t())
    tol = 0.02 * ref.abs().max().item() + 1e-3
    worst = 0.0
    for r in range(runs):
        got = ext.gemm_bt_256(A, B, None, False)
        worst = max(worst, maxerr(got, ref))
    return worst, tol


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--bench", action="store_true")
    args = ap.parse_args()
    ext = extension()

    ok = True
    # refcheck: odd/padded shapes + bias path
    for (M, N, K) in [(256, 256, 256), (512, 512, 512), (300, 200, 260),
                      (512, 256, 384), (256, 512, 1280)]:
        err, tol = refcheck(ext, M, N, K)
        good = err <= tol
        ok &= good
        print(json.dumps({"refcheck": [M, N, K], "max_err": round(err, 5),
                          "tol": round(tol, 5), "ok": good}))
    # A = I with asymmetric B: catches a transposed / misrouted C that
    # random-operand refchecks can miss (guide §5.4)
    I = torch.eye(256, device="cuda").bfloat16()
    Ba = torch.zeros(256, 256, device="cuda")
    Ba[3, 7] = 1.0
    Ba[200, 40] = -2.0
    got = ext.gemm_bt_256(I, Ba.bfloat16(), None, False).float()
    good = (got[7, 3].item() == 1.0 and got[40, 200].item() == -2.0
            and got.abs().sum().item() == 3.0)
    ok &= good
    print(json.dumps({"refcheck": "identity-asymB", "ok": good}))

    # bias
    torch.manual_seed(1)
    A = (torch.rand(256, 512, device="cuda") - 0.5).bfloat16()
    B = (torch.rand(256, 512, device="cuda") - 0.5).bfloat16()
    b = torch.randn(256, device="cuda")
    ref = A.float() @ B.float().t() + b
    err = maxerr(ext.gemm_bt_256(A, B, b, False), ref)
    good = err <= 0.02 * ref.abs().max().item() + 1e-3
    ok &= good
    print(json.dumps({"refcheck": "bias", "max_err": round(err, 5),
                      "ok": good}))

    # race screen: 30 repeated runs at three sizes, fresh data each
    for (M, N, K) in [(256, 256, 512), (512, 512, 1024), (4096, 4096, 4096)]:
        worst = 0.0
        for s in range(30 if M < 4096 else 10):
            err, tol = refcheck(ext, M, N, K, seed=s)
            worst = max(worst, err)
        good = worst <= tol
        ok &= good
        print(json.dumps({"race_screen": [M, N, K],
                          "worst_err": round(worst, 5), "ok": good}))

    if args.bench:
        for (M, N, K) in [(4096, 4096, 4096), (8192, 8192, 8192),
                          (50176, 384, 1152), (50176, 1152, 384)]:
            A = (torch.rand(M, K, device="cuda") - 0.5).bfloat16()
            B = (torch.rand(N, K, device="cuda") - 0.5).bfloat16()
            flop = 2.0 * M * N * K
            t256 = timeit(lambda: ext.gemm_bt_256(A, B, None, False))
            t128 = timeit(lambda: ext.gemm_bf16(A, B, False, False))
            Bt = B.t().contiguous()
            ttor = timeit(lambda: A @ Bt)
            print(json.dumps({"bench": [M, N, K],
                              "g256_TF": round(flop / t256 / 1e6, 1),
                              "g128_TF": round(flop / t128 / 1e6, 1),
                              "torch_TF": round(flop / ttor / 1e6, 1)}))

    print("PASS" if ok else "FAIL")
    sys.exit(0 if ok else 1)


if __name__ == "__main__":
    main()
