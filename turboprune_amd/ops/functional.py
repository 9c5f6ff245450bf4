"""Functional masked ops with reference-exact semantics.

The observable contract (reference: utils/mask_layers.py:23-34,59-70,
104-119): forward computes with ``mask * weight``; the autograd chain
therefore multiplies the weight gradient by the mask. Raw (unmasked)
weights keep receiving weight-decay/momentum updates — they are nullified
at forward time only — and checkpoints contain the raw weights.

MI355X design: ``mask_apply`` is a single fused HIP kernel producing the
compute-dtype masked weight (no fp32 temp + separate autocast cast), and
layers may hold a *cached* masked weight maintained by the fused SGD
kernel so steady-state forwards skip the multiply entirely.
"""

from __future__ import annotations

import os
from typing import Optional

import torch
import torch.nn.functional as F

from turboprune_amd.ops import _backend


def mask_apply(weight: torch.Tensor, mask: torch.Tensor,
               out_dtype: Optional[torch.dtype] = None) -> torch.Tensor:
    """out = (mask != 0) ? weight : 0, cast to out_dtype. No autograd."""
    out_dtype = out_dtype or weight.dtype
    if _backend.use_native(weight, mask):
        return _backend.extension().mask_apply(weight, mask, out_dtype)
    return (weight * mask).to(out_dtype)


def grad_mask_apply(grad: torch.Tensor, mask: torch.Tensor,
                    out_dtype: torch.dtype) -> torch.Tensor:
    """grad_weight = mask ⊙ grad, cast to the raw weight dtype."""
    if _backend.use_native(grad, mask):
        return _backend.extension().mask_apply(grad, mask, out_dtype)
    return (grad * mask.to(grad.dtype)).to(out_dtype)


class MaskedWeight(torch.autograd.Function):
    """Differentiable masked weight: forward yields ``mask ⊙ weight`` (or a
    pre-computed cache of it), backward applies the mask to the gradient and
    casts to the raw weight's dtype."""

    @staticmethod
    def forward(ctx, weight: torch.Tensor, mask: torch.Tensor,
                cache: Optional[torch.Tensor], compute_dtype: Optional[torch.dtype]):
        ctx.save_for_backward(mask)
        ctx.weight_dtype = weight.dtype
        if cache is not None:
            return cache
        return mask_apply(weight, mask, compute_dtype or weight.dtype)

    @staticmethod
    def backward(ctx, grad):
        (mask,) = ctx.saved_tensors
        return grad_mask_apply(grad, mask, ctx.weight_dtype), None, None, None


def masked_weight(weight: torch.Tensor, mask: torch.Tensor,
                  cache: Optional[torch.Tensor] = None,
                  compute_dtype: Optional[torch.dtype] = None) -> torch.Tensor:
    return MaskedWeight.apply(weight, mask, cache, compute_dtype)


def masked_conv2d(x, weight, mask, bias=None, stride=1, padding=0,
                  dilation=1, groups=1, cache=None, compute_dtype=None):
    w = masked_weight(weight, mask, cache, compute_dtype)
    return F.conv2d(x, w, bias, stride, padding, dilation, groups)


def masked_linear(x, weight, mask, bias=None, cache=None, compute_dtype=None):
    w = masked_weight(weight, mask, cache, compute_dtype)
    if w.dtype == torch.bfloat16 and x.dtype != w.dtype:
        # the bf16 masked-weight cache defines the compute dtype; cast
        # the activations like autocast would inside F.linear (also
        # required OUTSIDE autocast — F.linear raises on mixed dtypes)
        x = x.to(torch.bfloat16)
    # TURBOPRUNE_GEMM: auto (default, per-shape routing) | native
    # (always in-house MFMA GEMM) | library (always hipBLASLt)
    mode = os.environ.get("TURBOPRUNE_GEMM", "auto")
    if (mode != "library" and _backend.use_native(x, w) and x.dim() >= 2):
        ext = _backend.extension()
        if ext is not None and hasattr(ext, "masked_linear_available") \
                and ext.masked_linear_available(x, w) \
                and (mode == "native" or _gemm_shape_native_wins(x, w)):
            return _MaskedLinearGemm.apply(x, w, bias)
    return F.linear(x, w, bias)


def _gemm_shape_native_wins(x, w) -> bool:
    """Per-shape routing from the round-1 PMC table
    (profiles/r01_gemm_mfma_pmc.md): the in-house 128² MFMA GEMM beats
    hipBLASLt on tall token-GEMMs (M ≥ tens of thousands, N,K ≤ ~2k —
    every DeiT training shape, incl. its split-K grad_w at 2.1×) and
    loses on big squares (8192³: 901 vs 1565 TF) and on small-M
    classifier heads (ResNet50 fc 512×1000×2048: 48 vs 100 TF). Routing
    at the Function level is enough: when (M,N,K) is in the token
    regime, both backward GEMMs (M×K×N and the deep-K N×K×M) are in
    regimes ours wins too."""
    K = x.shape[-1]
    M = x.numel() // K
    N = w.shape[0]
    return M >= 4096 and N <= 2048 and K <= 2048


class _MaskedLinearGemm(torch.autograd.Function):
    """Linear on the in-house MFMA GEMM kernel (y = x @ w^T + b).

    Forward/backward both route through the HIP GEMM; backward grads:
      grad_x = grad_y @ w ; grad_w = grad_y^T @ x ; grad_b = sum(grad_y).
    """

    @staticmethod
    def forward(ctx, x, w, bias):
        ctx.save_for_backward(x, w)
        ctx.bias_dtype = bias.dtype if bias is not None else None
        ext = _backend.extension()
        return ext.linear_fwd(x, w, bias)

    @staticmethod
    def backward(ctx, grad_y):
        x, w = ctx.saved_tensors
        ext = _backend.extension()
        gy = grad_y.contiguous()
        grad_x, grad_w = ext.linear_bwd(gy, x, w)
        grad_b = None
        if ctx.bias_dtype is not None:
            gy2 = gy.reshape(-1, gy.shape[-1])
            if gy2.dtype == torch.bfloat16 and gy2.shape[-1] % 8 == 0:
                # one streaming HIP pass (the eager reduce was 5.5% of
                # the DeiT step, profiles/r02_bench_final.md)
                grad_b = ext.colsum_bf16(gy2).to(ctx.bias_dtype)
            else:
                grad_b = gy2.sum(0).to(ctx.bias_dtype)
        return grad_x, grad_w, grad_b


def synflow_linearize_(t: torch.Tensor):
    """t <- |t| in place; returns the sign tensor (reference
    pruning_utils.py:223-248). GPU: one fused HIP pass emitting int8
    signs; CPU oracle: torch sign + abs_."""
    if _backend.use_native(t) and t.dtype == torch.float32 \
            and t.is_contiguous():
        return _backend.extension().sign_abs_(t)
    s = torch.sign(t)
    t.abs_()
    return s


def synflow_restore_(t: torch.Tensor, sign: torch.Tensor):
    """t <- t * sign in place (signs from synflow_linearize_)."""
    if _backend.use_native(t) and t.dtype == torch.float32 \
            and sign.dtype == torch.int8 and t.is_contiguous():
        _backend.extension().mul_sign_(t, sign)
        return t
    return t.mul_(sign.to(t.dtype))


def bernoulli_mask_(mask: torch.Tensor, p: float,
                    seed: Optional[int] = None) -> torch.Tensor:
    """In-place Bernoulli(p) fill of a mask buffer (reference:
    mask_layers.py:43 ``zeros_like(weight).bernoulli_(p)``).

    On GPU a Philox counter-based HIP kernel is used (seeded from torch's
    RNG unless an explicit seed is given)."""
    if _backend.use_native(mask):
        if seed is None:
            seed = int(torch.randint(0, 2**31 - 1, (1,)).item())
        _backend.extension().bernoulli_mask_(mask, float(p), seed)
        return mask
    return mask.zero_().bernoulli_(p)


def kth_smallest(values: torch.Tensor, k: int) -> float:
    """k-th smallest (1-based) of a 1-D tensor — the pruning threshold
    primitive (reference: torch.kthvalue at utils/pruning_utils.py:79).

    GPU path: multi-pass radix select HIP kernel (no full sort)."""
    assert values.dim() == 1
    assert 1 <= k <= values.numel(), (k, values.numel())
    if _backend.use_native(values):
        return float(_backend.extension().kth_smallest(values, k))
    return float(torch.kthvalue(values.float().cpu(), k).values)


def mask_from_threshold_(mask: torch.Tensor, score: torch.Tensor,
                         threshold: float) -> torch.Tensor:
    """mask = where(score <= threshold, 0, 1) in place (reference:
    utils/pruning_utils.py:82-88)."""
    if _backend.use_native(mask, score):
        _backend.extension().mask_from_threshold_(mask, score, float(threshold))
        return mask
    mask.copy_(torch.where(score <= threshold,
                           torch.zeros_like(mask), torch.ones_like(mask)))
    return mask


def masked_abs_score(weight: torch.Tensor, mask: torch.Tensor,
                     other: Optional[torch.Tensor] = None) -> torch.Tensor:
    """|mask * weight| (magnitude) or |mask * weight * other| (SNIP/SynFlow
    style with other=grad). Reference: utils/pruning_utils.py:75,189-191."""
    if _backend.use_native(weight, mask):
        return _backend.extension().masked_abs_score(
            weight, mask, other if other is not None else torch.Tensor())
    s = (mask * weight).detach()
    if other is not None:
        s = s * other
    return s.abs()


def cross_entropy(logits: torch.Tensor, target: torch.Tensor) -> torch.Tensor:
    """Mean-reduction cross-entropy. GPU: fused log-softmax+NLL fwd/bwd HIP
    kernel (one pass over logits each way); CPU: torch reference."""
    if _backend.use_native(logits) and logits.dim() == 2:
        return _FusedCrossEntropy.apply(logits.contiguous(), target)
    return F.cross_entropy(logits, target)


class _FusedCrossEntropy(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, target):
        ext = _backend.extension()
        loss, lse = ext.ce_fwd(logits, target)
        ctx.save_for_backward(logits, target, lse)
        return loss

    @staticmethod
    def backward(ctx, grad_out):
        logits, target, lse = ctx.saved_tensors
        ext = _backend.extension()
        grad_logits = ext.ce_bwd(logits, target, lse, grad_out)
        return grad_logits, None


def accuracy_count(logits: torch.Tensor, target: torch.Tensor) -> torch.Tensor:
    """Number of argmax(logits)==target hits, as a device tensor (reference:
    torchmetrics Accuracy, base_harness.py:54-60 — synced per epoch here)."""
    if _backend.use_native(logits) and logits.dim() == 2:
        return _backend.extension().accuracy_count(logits.contiguous(), target)
    return (logits.argmax(dim=-1) == target).sum()
