"""Fused LayerNorm + exact GELU modules (ViT path). Drop-in for
nn.LayerNorm / nn.GELU; torch fallbacks on CPU and unsupported shapes
are the numerics oracles (see csrc/layernorm_gelu.hip)."""

from __future__ import annotations

import torch
import torch.nn as nn
import os

import torch.nn.functional as F

from turboprune_amd.ops import _backend


class _LNFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, gamma, beta, eps):
        ext = _backend.extension()
        y, mean, rstd = ext.ln_fwd(x, gamma, beta, eps)
        ctx.save_for_backward(x, gamma, mean, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, gamma, mean, rstd = ctx.saved_tensors
        ext = _backend.extension()
        dx, dgamma, dbeta = ext.ln_bwd(x, dy, gamma, mean, rstd)
        return dx, dgamma.to(gamma.dtype), dbeta.to(gamma.dtype), None


class FusedLayerNorm(nn.LayerNorm):
    """nn.LayerNorm with a fused HIP path for last-dim normalization of
    bf16/f32 tensors (state dict identical)."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        vn = 8 if x.dtype == torch.bfloat16 else 4
        if (os.environ.get("TURBOPRUNE_LN", "native") == "native"
                and x.is_cuda and len(self.normalized_shape) == 1
                and x.dtype in (torch.bfloat16, torch.float32)
                and x.shape[-1] == self.normalized_shape[0]
                and x.shape[-1] % vn == 0
                and self.weight is not None and self.bias is not None
                and _backend.use_native(x)):
            return _LNFn.apply(x, self.weight, self.bias, self.eps)
        return super().forward(x)


class _GeluFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ext = _backend.extension()
        ctx.save_for_backward(x)
        return ext.gelu_fwd(x)

    @staticmethod
    def backward(ctx, dy):
        (x,) = ctx.saved_tensors
        ext = _backend.extension()
        return ext.gelu_bwd(x, dy)


class FusedGELU(nn.Module):
    """Exact (erf) GELU with fused HIP fwd/bwd."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        vn = 8 if x.dtype == torch.bfloat16 else 4
        if (os.environ.get("TURBOPRUNE_GELU", "native") == "native"
                and x.is_cuda and x.dtype in (torch.bfloat16, torch.float32)
                and x.numel() % vn == 0 and _backend.use_native(x)):
            return _GeluFn.apply(x)
        return F.gelu(x)
