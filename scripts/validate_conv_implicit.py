#!/usr/bin/env python3
"""Numerics + timing for the experimental implicit-GEMM conv forward
vs MIOpen (F.conv2d), at ResNet50 shapes."""
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from turboprune_amd.ops._backend import extension  # noqa: E402

BS = int(os.environ.get("CONV_BS", "32"))
SHAPES = [  # (N, Cin, H, W, Cout, k, stride, pad)
    (BS, 64, 56, 56, 64, 3, 1, 1),
    (BS, 128, 28, 28, 128, 3, 1, 1),
    (BS, 256, 14, 14, 256, 3, 1, 1),
    (BS, 512, 7, 7, 512, 3, 1, 1),
    (BS, 128, 56, 56, 128, 3, 2, 1),   # stride-2
    (BS, 64, 56, 56, 256, 1, 1, 0),    # 1x1 via the same kernel
]


def timeit(fn, iters=10, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


def main():
    ext = extension()
    dev = "cuda:0"
    for (N, Cin, H, W, Cout, k, s, p) in SHAPES:
        torch.manual_seed(Cin + Cout)
        x = (torch.rand(N, Cin, H, W, device=dev) - 0.5).to(torch.bfloat16) \
            .to(memory_format=torch.channels_last)
        w = ((torch.rand(Cout, Cin, k, k, device=dev) - 0.5) * 0.1) \
            .to(torch.bfloat16).to(memory_format=torch.channels_last)
        y = ext.conv2d_implicit_fwd(x, w, None, s, p)
        ref = torch.nn.functional.conv2d(x, w, None, s, p)
        err = (y.float() - ref.float()).abs().max().item()
        scale = ref.float().abs().max().item()
        t_ours = timeit(lambda: ext.conv2d_implicit_fwd(x, w, None, s, p))
        t_miopen = timeit(lambda: torch.nn.functional.conv2d(x, w, None, s, p))
        flop = 2 * N * (H // s) * (W // s) * Cout * Cin * k * k
        print(json.dumps({
            "shape": [N, Cin, H, W, Cout, k, s, p],
            "max_err": round(err, 5), "ref_scale": round(scale, 2),
            "ok": err < 0.05 * max(scale, 1.0),
            "ours_us": round(t_ours, 1),
            "ours_TF": round(flop / t_ours / 1e6, 1),
            "miopen_us": round(t_miopen, 1),
            "miopen_TF": round(flop / t_miopen / 1e6, 1)}))




def grad_input_via_fwd(ext, gy, w, stride, pad):
    """grad_input for stride-1 conv = conv(gy, rot180(w)^T): the same
    implicit-GEMM forward kernel with a permuted weight."""
    assert stride == 1
    k = w.shape[2]
    w_rot = torch.flip(w, dims=[2, 3]).permute(1, 0, 2, 3) \
        .contiguous(memory_format=torch.channels_last)
    return ext.conv2d_implicit_fwd(gy.contiguous(
        memory_format=torch.channels_last), w_rot, None, 1, k - 1 - pad)


def validate_grad_input():
    ext = extension()
    dev = "cuda:0"
    for (N, Cin, H, W, Cout, k, s, p) in SHAPES:
        if s != 1 or Cout % 64 != 0:
            continue
        torch.manual_seed(Cin)
        x = (torch.rand(N, Cin, H, W, device=dev) - 0.5).to(torch.bfloat16) \
            .to(memory_format=torch.channels_last).requires_grad_()
        w = ((torch.rand(Cout, Cin, k, k, device=dev) - 0.5) * 0.1) \
            .to(torch.bfloat16).to(memory_format=torch.channels_last)
        y = torch.nn.functional.conv2d(x, w, None, s, p)
        gy = torch.randn_like(y)
        (ref,) = torch.autograd.grad(y, x, gy, retain_graph=True)
        got = grad_input_via_fwd(ext, gy, w, s, p)
        err = (got.float() - ref.float()).abs().max().item()
        scale = ref.float().abs().max().item()
        t_ours = timeit(lambda: grad_input_via_fwd(ext, gy, w, s, p))
        t_mi = timeit(lambda: torch.autograd.grad(
            y, x, gy, retain_graph=True), iters=5, warmup=2)
        print(json.dumps({"grad_input_shape": [N, Cin, H, W, Cout, k, s, p],
                          "max_err": round(err, 5),
                          "ok": err < 0.05 * max(scale, 1.0),
                          "ours_us": round(t_ours, 1),
                          "autograd_us": round(t_mi, 1)}))


if __name__ == "__main__":
    main()
    validate_grad_input()
