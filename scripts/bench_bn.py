#!/usr/bin/env python3
"""Microbenchmark: fused BN kernels vs MIOpen-composed at ResNet50 bs512
shapes. Prints per-shape times and effective GB/s."""

import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from turboprune_amd.ops._backend import extension  # noqa: E402

SHAPES = [(64, 112), (64, 56), (256, 56), (128, 56), (128, 28), (512, 28),
          (256, 28), (256, 14), (1024, 14), (512, 14), (512, 7), (2048, 7)]
N = 512


def timeit(fn, iters=10, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us


def main():
    ext = extension()
    dev = "cuda:0"
    results = []
    for C, HW in SHAPES:
        x = torch.randn(N, C, HW, HW, device=dev).to(torch.bfloat16) \
            .to(memory_format=torch.channels_last)
        res = torch.randn_like(x)
        dy = torch.randn_like(x)
        gamma = torch.ones(C, device=dev)
        beta = torch.zeros(C, device=dev)
        rm = torch.zeros(C, device=dev)
        rv = torch.ones(C, device=dev)

        def fused_fwd():
            return ext.bn_fwd(x, res, gamma, beta, rm, rv, True, 0.1,
                              1e-5, True)

        y, mean, rstd, rmask = fused_fwd()

        def fused_bwd():
            return ext.bn_bwd(x, rmask, dy, gamma, mean, rstd, True, True)

        def miopen_fwd():
            out = torch.nn.functional.batch_norm(
                x, rm, rv, gamma, beta, True, 0.1, 1e-5)
            return torch.relu(out + res)

        xg = x.detach().clone().requires_grad_()
        rg = res.detach().clone().requires_grad_()
        out_ref = torch.relu(torch.nn.functional.batch_norm(
            xg, rm, rv, gamma, beta, True, 0.1, 1e-5) + rg)

        def miopen_bwd():
            g = torch.autograd.grad(out_ref, [xg, rg], dy,
                                    retain_graph=True)
            return g

        t_ff = timeit(fused_fwd)
        t_fb = timeit(fused_bwd)
        skip_ref = os.environ.get("BN_SKIP_MIOPEN") == "1"
        t_mf = 0.0 if skip_ref else timeit(miopen_fwd)
        t_mb = 0.0 if skip_ref else timeit(miopen_bwd)
        # device copy-bandwidth reference at this tensor size: the
        # streaming ceiling the apply passes should approach
        t_cp = timeit(lambda: dy.copy_(x))

        nbytes = N * C * HW * HW * 2
        # fwd: read x,res write y (+ reduce read x) = 4 passes
        gbs_ff = 4 * nbytes / t_ff / 1e3
        # bwd: reduce reads x,y,dy; apply reads x,y,dy writes dx,dres = 8
        gbs_fb = 8 * nbytes / t_fb / 1e3
        cp_GBs = 2 * nbytes / t_cp / 1e3
        row = dict(C=C, HW=HW, MB=round(nbytes / 1e6, 1),
                   copy_GBs=round(cp_GBs),
                   fused_fwd_us=round(t_ff, 1), miopen_fwd_us=round(t_mf, 1),
                   fused_bwd_us=round(t_fb, 1), miopen_bwd_us=round(t_mb, 1),
                   fwd_GBs=round(gbs_ff), bwd_GBs=round(gbs_fb))
        results.append(row)
        print(json.dumps(row))
    tot_ff = sum(r["fused_fwd_us"] for r in results)
    tot_mf = sum(r["miopen_fwd_us"] for r in results)
    tot_fb = sum(r["fused_bwd_us"] for r in results)
    tot_mb = sum(r["miopen_bwd_us"] for r in results)
    print(json.dumps(dict(total_fused_fwd=tot_ff, total_miopen_fwd=tot_mf,
                          total_fused_bwd=tot_fb, total_miopen_bwd=tot_mb)))


if __name__ == "__main__":
    main()
