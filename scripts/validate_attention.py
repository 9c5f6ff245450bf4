"""On-device validation for the EXPERIMENTAL fused attention forward
(csrc/attention.hip). Run on an MI355X box:

    python scripts/validate_attention.py          # refcheck + race screen
    python scripts/validate_attention.py --bench  # + A/B vs torch SDPA

Random operands (zero-filled Q/K collapse softmax work — guide §5.4
rule 25); repeated runs because sync bugs are timing-dependent."""

import argparse
import json
import math
import sys
import time

import torch

sys.path.insert(0, ".")
from turboprune_amd.ops._backend import extension  # noqa: E402


def ref_attn(q, k, v, scale):
    s = (q.float() @ k.float().transpose(-2, -1)) * scale
    return torch.softmax(s, dim=-1) @ v.float()


def check(ext, B, H, S, runs=1, seed=0):
    torch.manual_seed(seed)
    mk = lambda: (torch.rand(B, H, S, 64, device="cuda") - 0.5).bfloat16()
    q, k, v = mk(), mk(), mk()
    scale = 1.0 / math.sqrt(64)
    ref = ref_attn(q, k, v, scale)
    worst = 0.0
    for _ in range(runs):
        got, lse = ext.attn_fwd(q, k, v, scale)
        worst = max(worst, (got.float() - ref).abs().max().item())
        # lse check: logsumexp of the scaled scores
        lref = torch.logsumexp(
            (q.float() @ k.float().transpose(-2, -1)) * scale, dim=-1)
        worst = max(worst, (lse - lref).abs().max().item() * 0.1)
    return worst


def timeit(fn, iters=30, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--bench", action="store_true")
    args = ap.parse_args()
    ext = extension()
    ok = True
    # tol: bf16 P + f32 accum; outputs are O(1)
    for (B, H, S) in [(1, 1, 64), (2, 3, 197), (4, 6, 100), (2, 2, 1024)]:
        err = check(ext, B, H, S)
        good = err < 0.03
        ok &= good
        print(json.dumps({"refcheck": [B, H, S], "max_err": round(err, 5),
                          "ok": good}))
    for (B, H, S) in [(2, 3, 197), (8, 6, 512)]:
        worst = max(check(ext, B, H, S, runs=5, seed=s) for s in range(6))
        good = worst < 0.03
        ok &= good
        print(json.dumps({"race_screen": [B, H, S],
                          "worst_err": round(worst, 5), "ok": good}))

    # backward: fused kernel vs autograd through explicit attention
    for (B, H, S) in [(1, 1, 64), (2, 3, 197)]:
        torch.manual_seed(10 + S)
        mk = lambda: ((torch.rand(B, H, S, 64, device="cuda") - 0.5)
                      .bfloat16().requires_grad_())
        q, k, v = mk(), mk(), mk()
        scale = 1.0 / math.sqrt(64)
        out, lse = ext.attn_fwd(q.detach(), k.detach(), v.detach(), scale)
        s_ref = (q.float() @ k.float().transpose(-2, -1)) * scale
        o_ref = torch.softmax(s_ref, dim=-1) @ v.float()
        do = torch.randn_like(o_ref)
        o_ref.backward(do)
        dq, dk, dv = ext.attn_bwd(q.detach(), k.detach(), v.detach(), out,
                                  do.bfloat16(), lse, scale)
        errs = [
            (dq.float() - q.grad.float()).abs().max().item()
            / (q.grad.float().abs().max().item() + 1.0),
            (dk - k.grad.float()).abs().max().item()
            / (k.grad.float().abs().max().item() + 1.0),
            (dv - v.grad.float()).abs().max().item()
            / (v.grad.float().abs().max().item() + 1.0)]
        good = max(errs) < 0.05  # relative to grad magnitude (bf16 P)
        ok &= good
        print(json.dumps({"bwd": [B, H, S],
                          "max_err_dq_dk_dv": [round(e, 4) for e in errs],
                          "ok": good}))

    if args.bench:
        for (B, H, S) in [(256, 6, 197), (64, 12, 197), (32, 6, 1024)]:
            mk = lambda: (torch.rand(B, H, S, 64, device="cuda") - 0.5) \
                .bfloat16()
            q, k, v = mk(), mk(), mk()
            scale = 1.0 / math.sqrt(64)
            t_ours = timeit(lambda: ext.attn_fwd(q, k, v, scale))
            t_sdpa = timeit(
                lambda: torch.nn.functional
                .scaled_dot_product_attention(q, k, v))
            t_bwd = None
            try:
                out, lse = ext.attn_fwd(q, k, v, scale)
                do = torch.randn_like(out)
                t_bwd = round(timeit(lambda: ext.attn_bwd(
                    q, k, v, out, do, lse, scale)), 1)
            except Exception as e:  # noqa: BLE001
                t_bwd = repr(e)
            flop = 4.0 * B * H * S * S * 64
            print(json.dumps({"bench": [B, H, S],
                              "ours_us": round(t_ours, 1),
                              "ours_TF": round(flop / t_ours / 1e6, 1),
                              "sdpa_us": round(t_sdpa, 1),
                              "bwd_us": t_bwd}))

    print("PASS" if ok else "FAIL")
    sys.exit(0 if ok else 1)


if __name__ == "__main__":
    main()
