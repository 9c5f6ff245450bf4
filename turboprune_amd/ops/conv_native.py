"""Full in-house conv triple (fwd / grad_input / grad_weight) as one
autograd Function over a pluggable backend.

The backend supplies two primitives: a forward conv and a weight-grad
(wrw). grad_input is NOT a primitive — it is the forward conv of the
(zero-dilated) output gradient with the rotated-transposed weight
(``ops/conv_backward.py``), so the whole backward runs on the same two
kernels. ``TorchBackend`` (F.conv2d / torch.nn.grad) lets every piece of
this wiring be tested on CPU in fp32; ``NativeBackend`` swaps in the
validated implicit-GEMM MFMA kernels (csrc/conv_implicit.hip,
csrc/conv_wrw.hip) without touching the Function.

Opt-in on GPU via ``TURBOPRUNE_CONV=native`` (ConvMask checks
``native_conv_ok``); the default path remains the library conv until the
dispatch is GPU-tuned per shape (docs/ROADMAP_ROUND2.md item 1).
"""

from __future__ import annotations

import os
from typing import Optional

import torch
import torch.nn.functional as F

from turboprune_amd.ops.conv_backward import conv_grad_input


class TorchBackend:
    """CPU/fp32 oracle backend."""

    @staticmethod
    def fwd(x, w, bias, stride, pad):
        return F.conv2d(x, w, bias, stride, pad)

    @staticmethod
    def gradin(gy, w, in_hw, stride, pad):
        return conv_grad_input(gy, w, in_hw, stride, pad,
                               conv_fn=lambda a, b, s, p:
                               F.conv2d(a, b, None, s, p))

    @staticmethod
    def wrw(gy, x, w_shape, stride, pad):
        return torch.nn.grad.conv2d_weight(x, w_shape, gy, stride, pad)


class NativeBackend:
    """MFMA implicit-GEMM kernels (bf16 NHWC, gfx950)."""

    @staticmethod
    def fwd(x, w, bias, stride, pad):
        from turboprune_amd.ops._backend import extension
        return extension().conv2d_implicit_fwd(x, w, bias, stride, pad)

    @staticmethod
    def gradin(gy, w, in_hw, stride, pad):
        # one fused kernel: rotated-weight forward conv with the
        # zero-inserted gy coordinates resolved inside the im2col gather
        from turboprune_amd.ops._backend import extension
        return extension().conv2d_implicit_gradin(
            gy, w, in_hw[0], in_hw[1], stride, pad)

    @staticmethod
    def wrw(gy, x, w_shape, stride, pad):
        from turboprune_amd.ops._backend import extension
        return extension().conv2d_implicit_wrw(
            gy, x, w_shape[2], w_shape[3], stride, pad)


class ConvImplicitFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, bias, stride, pad, backend):
        ctx.save_for_backward(x, w)
        ctx.stride, ctx.pad, ctx.backend = stride, pad, backend
        ctx.bias_dtype = None if bias is None else bias.dtype
        return backend.fwd(x, w, bias, stride, pad)

    @staticmethod
    def backward(ctx, gy):
        x, w = ctx.saved_tensors
        stride, pad, backend = ctx.stride, ctx.pad, ctx.backend
        gx = gw = gb = None
        if ctx.needs_input_grad[0]:
            gx = backend.gradin(gy, w, (x.shape[2], x.shape[3]), stride,
                                pad)
        if ctx.needs_input_grad[1]:
            gw = backend.wrw(gy, x, w.shape, stride, pad)
            if gw.dtype != w.dtype:
                gw = gw.to(w.dtype)
        if ctx.bias_dtype is not None and ctx.needs_input_grad[2]:
            gb = gy.float().sum(dim=(0, 2, 3)).to(ctx.bias_dtype)
        return gx, gw, gb, None, None, None


def conv2d(x: torch.Tensor, w: torch.Tensor, bias: Optional[torch.Tensor],
           stride: int, pad: int, backend=None) -> torch.Tensor:
    return ConvImplicitFn.apply(x, w, bias, stride, pad,
                                backend or TorchBackend)


class MaskedConvNativeFn(torch.autograd.Function):
    """Masked Conv2d as ONE autograd node from the fp32 master weight
    (reference semantics: utils/mask_layers.py:25-34 under autocast).

    Forward computes on the bf16 cached masked weight (zero per-forward
    mask multiplies); backward delivers ``grad_weight = mask ⊙ wrw`` in
    fp32 straight to the master weight — no bf16 rounding at the
    Function boundary (the engine would silently downcast any fp32 grad
    returned for a bf16 input, so the conv and the mask-apply must live
    in the same node)."""

    @staticmethod
    def forward(ctx, x, weight, mask, bias, stride, pad, w_c, backend):
        ctx.save_for_backward(x, mask, w_c)
        ctx.stride, ctx.pad, ctx.backend = stride, pad, backend
        ctx.bias_dtype = None if bias is None else bias.dtype
        ctx.weight_dtype = weight.dtype
        return backend.fwd(x, w_c, bias, stride, pad)

    @staticmethod
    def backward(ctx, gy):
        from turboprune_amd.ops import functional as TF
        x, mask, w_c = ctx.saved_tensors
        stride, pad, backend = ctx.stride, ctx.pad, ctx.backend
        gx = gw = gb = None
        if ctx.needs_input_grad[0]:
            gx = backend.gradin(gy, w_c, (x.shape[2], x.shape[3]),
                                stride, pad)
        if ctx.needs_input_grad[1]:
            gw = backend.wrw(gy, x, w_c.shape, stride, pad)
            gw = TF.grad_mask_apply(gw, mask, ctx.weight_dtype)
        if ctx.bias_dtype is not None and ctx.needs_input_grad[3]:
            gb = gy.float().sum(dim=(0, 2, 3)).to(ctx.bias_dtype)
        return gx, gw, None, gb, None, None, None, None


def masked_conv2d_native(x, weight, mask, bias, stride: int, pad: int,
                         w_cache: torch.Tensor) -> torch.Tensor:
    return MaskedConvNativeFn.apply(x, weight, mask, bias, stride, pad,
                                    w_cache, NativeBackend)


def shape_ok(cout: int, cin: int, k: int, kw: int, stride, padding,
             dilation, groups: int) -> bool:
    """Shape half of the dispatch envelope: square k∈{1,3} with
    canonical padding, stride 1/2, both channel counts %64 (Cin for the
    fwd/wrw tiles, Cout because grad_input re-enters the forward with
    gy's channels as the contraction). On ResNet50 this admits every
    conv except the Cin=3 stem (tests/test_conv_native_fn.py pins it)."""
    if groups != 1 or dilation != (1, 1):
        return False
    if kw != k or k not in (1, 3):
        return False
    if stride[0] != stride[1] or stride[0] not in (1, 2):
        return False
    if padding != (k // 2, k // 2):
        return False
    return cout % 64 == 0 and cin % 64 == 0


def native_conv_ok(x: torch.Tensor, w: torch.Tensor, stride, padding,
                   dilation, groups: int) -> bool:
    """Full gate: opt-in env + bf16 channels_last GPU tensors +
    shape_ok."""
    if os.environ.get("TURBOPRUNE_CONV", "") != "native":
        return False
    if not (x.is_cuda and x.dtype == torch.bfloat16
            and w.dtype == torch.bfloat16):
        return False
    if not shape_ok(w.shape[0], w.shape[1], w.shape[2], w.shape[3],
                    stride, padding, dilation, groups):
        return False
    return x.is_contiguous(memory_format=torch.channels_last)
