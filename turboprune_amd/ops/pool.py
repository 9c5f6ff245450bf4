"""Fused NHWC MaxPool2d (see csrc/maxpool.hip). Drop-in for
nn.MaxPool2d (no state); falls back to torch on CPU / unsupported
shapes. Backward gathers via the forward's uint8 argmax map."""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from turboprune_amd.ops import _backend


class _MaxPoolFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, kh, kw, sh, sw, ph, pw):
        ext = _backend.extension()
        y, idx = ext.maxpool_fwd(x, kh, kw, sh, sw, ph, pw)
        ctx.save_for_backward(idx)
        ctx.meta = (x.shape[0], x.shape[1], x.shape[2], x.shape[3],
                    kh, kw, sh, sw, ph, pw)
        return y

    @staticmethod
    def backward(ctx, dy):
        (idx,) = ctx.saved_tensors
        n, c, hi, wi, kh, kw, sh, sw, ph, pw = ctx.meta
        ext = _backend.extension()
        dx = ext.maxpool_bwd(dy, idx, n, c, hi, wi, kh, kw, sh, sw, ph, pw)
        return dx, None, None, None, None, None, None


def _pair(v):
    return (v, v) if isinstance(v, int) else tuple(v)


class FusedMaxPool2d(nn.Module):
    def __init__(self, kernel_size, stride=None, padding=0):
        super().__init__()
        self.kernel_size = _pair(kernel_size)
        self.stride = _pair(stride if stride is not None else kernel_size)
        self.padding = _pair(padding)

    def extra_repr(self):
        return (f"kernel_size={self.kernel_size}, stride={self.stride}, "
                f"padding={self.padding}")

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        kh, kw = self.kernel_size
        sh, sw = self.stride
        ph, pw = self.padding
        vn = 8 if x.dtype == torch.bfloat16 else 4
        if (x.is_cuda and x.dim() == 4
                and x.is_contiguous(memory_format=torch.channels_last)
                and x.dtype in (torch.bfloat16, torch.float32)
                and x.shape[1] % vn == 0
                and _backend.use_native(x)):
            return _MaxPoolFn.apply(x, kh, kw, sh, sw, ph, pw)
        return F.max_pool2d(x, self.kernel_size, self.stride, self.padding)
