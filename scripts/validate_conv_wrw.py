#!/usr/bin/env python3
"""One-shot numerics check of the experimental wrw kernel."""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from turboprune_amd.ops._backend import extension  # noqa: E402

ext = extension()
dev = "cuda:0"
for (N, Cin, H, W, Cout, k, s, p) in [(8, 64, 28, 28, 64, 3, 1, 1),
                                      (8, 64, 28, 28, 128, 3, 2, 1),
                                      (4, 64, 16, 16, 128, 1, 1, 0)]:
    torch.manual_seed(Cin + Cout + s)
    x = (torch.rand(N, Cin, H, W, device=dev) - 0.5).to(torch.bfloat16) \
        .to(memory_format=torch.channels_last).requires_grad_()
    w = ((torch.rand(Cout, Cin, k, k, device=dev) - 0.5) * 0.1) \
        .to(torch.bfloat16).to(memory_format=torch.channels_last) \
        .requires_grad_()
    y = torch.nn.functional.conv2d(x, w, None, s, p)
    gy = torch.randn_like(y)
    (ref,) = torch.autograd.grad(y, w, gy)
    got = ext.conv2d_implicit_wrw(gy, x.detach(), k, k, s, p)
    err = (got.float() - ref.float()).abs().max().item()
    scale = ref.float().abs().max().item()
    print({"shape": (N, Cin, H, W, Cout, k, s, p),
           "max_err": round(err, 4), "scale": round(scale, 2),
           "ok": err < 0.05 * max(scale, 1.0)})
