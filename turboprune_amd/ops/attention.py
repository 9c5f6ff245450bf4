"""Scaled-dot-product attention for the DeiT path.

Default is torch SDPA, BY MEASUREMENT: the fused flash-style HIP
forward (csrc/attention.hip: online softmax, one block per
(batch*head, 64-q-row tile), head_dim 64) is numerically validated on
device (scripts/validate_attention.py, r2 checks) but runs 182 µs vs
AOTriton SDPA's 90 µs on the DeiT shape and costs −12% DeiT e2e, so it
stays opt-in via ``TURBOPRUNE_ATTN=native`` until tuned
(docs/ROADMAP_ROUND3.md item 3). SURVEY §5 allows SDPA to remain
torch-ROCm for this workload.

Backward is closed-form recompute (the flash-attention backward
identities), expressed in torch ops so it works with either forward and
is CPU-testable exactly:

    dV = P^T dO
    dP = dO V^T
    dS = P * (dP - rowsum(dP * P))      (softmax Jacobian)
    dQ = scale * dS K,  dK = scale * dS^T Q
"""

from __future__ import annotations

import math
import os

import torch
import torch.nn.functional as F


class _FlashAttn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, scale):
        from turboprune_amd.ops._backend import extension
        out, lse = extension().attn_fwd(q, k, v, scale)
        ctx.save_for_backward(q, k, v, out, lse)
        ctx.scale = scale
        return out

    @staticmethod
    def backward(ctx, do):
        from turboprune_amd.ops._backend import extension
        q, k, v, out, lse = ctx.saved_tensors
        if os.environ.get("TURBOPRUNE_ATTN_BWD", "native") == "torch":
            gq, gk, gv = attn_backward(q, k, v, do, ctx.scale)
        else:
            dq, dk, dv = extension().attn_bwd(q, k, v, out,
                                              do.contiguous(), lse,
                                              ctx.scale)
            gq, gk, gv = dq, dk.to(k.dtype), dv.to(v.dtype)
        return gq, gk, gv, None


def attn_backward(q, k, v, do, scale):
    """Recompute-based attention backward (fp32 internally)."""
    qf, kf, vf, dof = (t.float() for t in (q, k, v, do))
    s = (qf @ kf.transpose(-2, -1)) * scale
    p = torch.softmax(s, dim=-1)
    dv = p.transpose(-2, -1) @ dof
    dp = dof @ vf.transpose(-2, -1)
    ds = p * (dp - (dp * p).sum(dim=-1, keepdim=True))
    dq = (ds @ kf) * scale
    dk = (ds.transpose(-2, -1) @ qf) * scale
    return dq.to(q.dtype), dk.to(k.dtype), dv.to(v.dtype)


def _native_ok(q: torch.Tensor) -> bool:
    return (os.environ.get("TURBOPRUNE_ATTN", "") == "native"
            and q.is_cuda and q.dtype == torch.bfloat16
            and q.dim() == 4 and q.shape[-1] == 64)


def sdpa(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor) -> torch.Tensor:
    """(B, H, S, D) attention, non-causal, no dropout."""
    if _native_ok(q):
        scale = 1.0 / math.sqrt(q.shape[-1])
        return _FlashAttn.apply(q, k, v, scale)
    return F.scaled_dot_product_attention(q, k, v)
