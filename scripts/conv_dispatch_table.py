#!/usr/bin/env python3
"""Per-shape, per-op A/B of the native implicit-GEMM conv triple vs the
library (MIOpen) at the bench geometry — the measurement behind
ops/conv_native.py's dispatch table (promote by
measurement, not by flipping the flag blind).

For every unique ConvMask geometry in the flagship model (extracted by a
CPU shape-trace forward), times:

  fwd:    ext.conv2d_implicit_fwd        vs  F.conv2d
  gradin: ext.conv2d_implicit_gradin     vs  aten.convolution_backward[0]
  wrw:    ext.conv2d_implicit_wrw        vs  aten.convolution_backward[1]

and prints one JSON line per (shape, op) with µs + winner, then a
summary table aggregated by step share. Writes
gpurun_out/conv_table.json for the dispatch table generator.
"""
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402
import torch.nn.functional as F  # noqa: E402


def model_conv_shapes(model_name="resnet50", image=224):
    """(Cin, Cout, k, s, Hin, count) for every ConvMask, via CPU trace."""
    from turboprune_amd.config import compose
    from turboprune_amd.models import build_model
    from turboprune_amd.ops.mask_layers import ConvMask

    cfg = compose("bench_resnet50_imagenet",
                  [f"model_params.model_name={model_name}"])
    pm = build_model(cfg)
    shapes = {}
    hooks = []

    def hook(m, inp, out):
        k = (m.in_channels, m.out_channels, m.kernel_size[0], m.stride[0],
             inp[0].shape[-1])
        shapes[k] = shapes.get(k, 0) + 1

    for m in pm.modules():
        if isinstance(m, ConvMask):
            hooks.append(m.register_forward_hook(hook))
    with torch.no_grad():
        pm(torch.randn(1, 3, image, image))
    for h in hooks:
        h.remove()
    return shapes


def timeit(fn, iters=10, warmup=3):
    import time
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


def main():
    from turboprune_amd.ops._backend import extension
    ext = extension()
    dev = "cuda:0"
    bs = int(os.environ.get("CONV_BS", "512"))
    shapes = model_conv_shapes()
    rows = []
    for (cin, cout, k, s, hi), count in sorted(shapes.items()):
        if cin % 64 or cout % 64 or k not in (1, 3):
            print(json.dumps({"shape": [cin, cout, k, s, hi],
                              "count": count, "skip": "outside envelope"}))
            continue
        p = k // 2
        torch.manual_seed(cin + cout + s)
        x = (torch.rand(bs, cin, hi, hi, device=dev) - .5).bfloat16() \
            .to(memory_format=torch.channels_last)
        w = ((torch.rand(cout, cin, k, k, device=dev) - .5) * .1) \
            .bfloat16().to(memory_format=torch.channels_last)
        ho = (hi + 2 * p - k) // s + 1
        gy = torch.randn(bs, cout, ho, ho, device=dev).bfloat16() \
            .to(memory_format=torch.channels_last)

        flop = 2 * bs * ho * ho * cout * cin * k * k
        row = {"shape": [cin, cout, k, s, hi], "count": count,
               "flop_G": round(flop / 1e9, 1)}

        # forward
        t_n = timeit(lambda: ext.conv2d_implicit_fwd(x, w, None, s, p))
        t_l = timeit(lambda: F.conv2d(x, w, None, s, p))
        row["fwd"] = {"native_us": round(t_n, 1), "lib_us": round(t_l, 1)}

        # grad_input through NativeBackend (parity sub-convs for s2)
        from turboprune_amd.ops.conv_native import NativeBackend
        t_n = timeit(lambda: NativeBackend.gradin(gy, w, (hi, hi), s, p))
        t_l = timeit(lambda: torch.ops.aten.convolution_backward(
            gy, x, w, None, (s, s), (p, p), (1, 1), False, (0, 0), 1,
            (True, False, False))[0])
        row["gradin"] = {"native_us": round(t_n, 1), "lib_us": round(t_l, 1)}

        # wrw
        t_n = timeit(lambda: ext.conv2d_implicit_wrw(gy, x, k, k, s, p))
        t_l = timeit(lambda: torch.ops.aten.convolution_backward(
            gy, x, w, None, (s, s), (p, p), (1, 1), False, (0, 0), 1,
            (False, True, False))[1])
        row["wrw"] = {"native_us": round(t_n, 1), "lib_us": round(t_l, 1)}

        for op in ("fwd", "gradin", "wrw"):
            d = row[op]
            d["winner"] = "native" if d["native_us"] <= d["lib_us"] else "lib"
            d["ratio"] = round(d["native_us"] / d["lib_us"], 3)
        print(json.dumps(row), flush=True)
        rows.append(row)

    # aggregate: time per step if always-native vs always-lib vs per-op best
    tot = {"native": 0.0, "lib": 0.0, "best": 0.0}
    for r in rows:
        for op in ("fwd", "gradin", "wrw"):
            d = r[op]
            tot["native"] += d["native_us"] * r["count"]
            tot["lib"] += d["lib_us"] * r["count"]
            tot["best"] += min(d["native_us"], d["lib_us"]) * r["count"]
    print(json.dumps({"per_step_conv_us": {k: round(v, 0)
                                           for k, v in tot.items()}}))
    os.makedirs("gpurun_out", exist_ok=True)
    with open("gpurun_out/conv_table.json", "w") as f:
        json.dump({"bs": bs, "rows": rows, "totals": tot}, f, indent=1)


if __name__ == "__main__":
    main()
