#!/usr/bin/env python3
"""MFMA masked-GEMM microbench: TF/s at DeiT/ResNet shapes + square
sizes, fwd and bwd. Used with rocprofv3 --pmc for MFMA-utilization
evidence (profiles/)."""

import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from turboprune_amd.ops._backend import extension  # noqa: E402

# (M, N, K) — DeiT-S bs256 qkv / proj / fc1 / fc2, ResNet50 fc, squares
SHAPES = [
    (50432, 1152, 384),   # deit qkv fwd
    (50432, 384, 1152),   # deit proj / qkv grad_x
    (50432, 1536, 384),   # deit fc1
    (50432, 384, 1536),   # deit fc2
    (1152, 384, 50432),   # deit grad_w (split-K shape)
    (512, 1000, 2048),    # resnet50 fc
    (4096, 4096, 4096),   # square reference
    (8192, 8192, 8192),
]


def timeit(fn, iters=10, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    ext = extension()
    dev = "cuda:0"
    for M, N, K in SHAPES:
        torch.manual_seed(0)
        A = (torch.rand(M, K, device=dev) * 2 - 1).to(torch.bfloat16)
        B = (torch.rand(N, K, device=dev) * 2 - 1).to(torch.bfloat16)

        t = timeit(lambda: ext.gemm_bf16(A, B, False, False))
        tf = 2.0 * M * N * K / t / 1e12
        t_lib = timeit(lambda: A @ B.t())
        tf_lib = 2.0 * M * N * K / t_lib / 1e12
        print(json.dumps({"M": M, "N": N, "K": K,
                          "ours_us": round(t * 1e6, 1),
                          "ours_TF": round(tf, 1),
                          "hipblaslt_us": round(t_lib * 1e6, 1),
                          "hipblaslt_TF": round(tf_lib, 1)}))


if __name__ == "__main__":
    main()
