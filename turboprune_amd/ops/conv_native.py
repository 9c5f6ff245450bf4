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
        from turboprune_amd.ops._backend import extension
        if stride == 2 and os.environ.get("TURBOPRUNE_GRADIN_S2",
                                          "dilated") == "parity":
            # four dense stride-1 sub-convs; measured r2j: the eager
            # pad + strided-scatter glue makes this SLOWER than the
            # fused dilated kernel on device (1.0-1.7 ms vs 0.8-0.9),
            # so it stays opt-in until the scatter is fused into the
            # conv epilogue; see conv_backward.py
            from turboprune_amd.ops.conv_backward import \
                conv_grad_input_s2_parity
            ext = extension()
            return conv_grad_input_s2_parity(
                gy, w, in_hw, pad,
                conv_fn=lambda a, b, s, p:
                ext.conv2d_implicit_fwd(a, b, None, s, p))
        # fused kernel: rotated-weight forward conv with the
        # zero-inserted gy coordinates resolved inside the im2col gather
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


class LibBackend:
    """Library (MIOpen) primitives with the same interface — used by the
    per-shape dispatch for the ops where MIOpen wins."""

    @staticmethod
    def fwd(x, w, bias, stride, pad):
        return F.conv2d(x, w, bias, stride, pad)

    @staticmethod
    def gradin(gy, w, in_hw, stride, pad):
        n, _, _, _ = gy.shape
        size = (n, w.shape[1], in_hw[0], in_hw[1])
        return torch.nn.grad.conv2d_input(size, w, gy, stride, pad)

    @staticmethod
    def wrw(gy, x, w_shape, stride, pad):
        return torch.nn.grad.conv2d_weight(x, w_shape, gy, stride, pad)

    @staticmethod
    def both(gy, x, w, stride, pad, need_gx, need_gw):
        """One fused convolution_backward computing both grads (what the
        library autograd path would do — cheaper than two calls)."""
        gx, gw, _ = torch.ops.aten.convolution_backward(
            gy, x, w, None, (stride, stride), (pad, pad), (1, 1), False,
            (0, 0), 1, (need_gx, need_gw, False))
        return gx, gw


class AutoBackend:
    """Per-(shape, op) dispatch between the native implicit-GEMM kernels
    and MIOpen, driven by the measured table (scripts/
    conv_dispatch_table.py at bs512 on MI355X; see profiles/). Keys are
    (Cin, Cout, k, stride); unlisted shapes default to whichever side
    ``DEFAULT_NATIVE`` names per op."""

    # filled by _load_table() from _CONV_TABLE below (or the JSON file
    # named by TURBOPRUNE_CONV_TABLE, for A/B without rebuilds)
    table: dict = {}
    DEFAULT_NATIVE = {"fwd": True, "gradin": True, "wrw": False}

    @classmethod
    def _ops_for(cls, key):
        return cls.table.get(key, cls.DEFAULT_NATIVE)

    @staticmethod
    def fwd(x, w, bias, stride, pad):
        key = (w.shape[1], w.shape[0], w.shape[2], stride)
        use = AutoBackend._ops_for(key).get("fwd", True)
        return (NativeBackend if use else LibBackend).fwd(
            x, w, bias, stride, pad)

    @staticmethod
    def gradin(gy, w, in_hw, stride, pad):
        # note: w here is already the rotated layout consumer inside
        # NativeBackend; key on the FORWARD geometry
        key = (w.shape[1], w.shape[0], w.shape[2], stride)
        use = AutoBackend._ops_for(key).get("gradin", True)
        return (NativeBackend if use else LibBackend).gradin(
            gy, w, in_hw, stride, pad)

    @staticmethod
    def wrw(gy, x, w_shape, stride, pad):
        key = (w_shape[1], w_shape[0], w_shape[2], stride)
        use = AutoBackend._ops_for(key).get("wrw", False)
        return (NativeBackend if use else LibBackend).wrw(
            gy, x, w_shape, stride, pad)

    @staticmethod
    def both(gy, x, w, stride, pad, need_gx, need_gw):
        key = (w.shape[1], w.shape[0], w.shape[2], stride)
        plan = AutoBackend._ops_for(key)
        if plan.get("gradin", True) or plan.get("wrw", False):
            return None  # at least one native op: per-op path
        return LibBackend.both(gy, x, w, stride, pad, need_gx, need_gw)

    @staticmethod
    def any_native(cout, cin, k, stride) -> bool:
        plan = AutoBackend._ops_for((cin, cout, k, stride))
        return bool(plan.get("fwd", True) or plan.get("gradin", True)
                    or plan.get("wrw", False))


def _load_table() -> None:
    import json
    path = os.environ.get("TURBOPRUNE_CONV_TABLE", "")
    if path and os.path.exists(path):
        with open(path) as f:
            raw = json.load(f)
        AutoBackend.table = {tuple(json.loads(k)): v
                             for k, v in raw.items()}
    else:
        AutoBackend.table = dict(_CONV_TABLE)


# Measured dispatch table — scripts/conv_dispatch_table.py at bs512 on
# MI355X (profiles/r02_conv_dispatch.md). Keys (Cin, Cout, k, stride);
# True = the native implicit-GEMM kernel beat MIOpen for that op at
# that shape. Current kernel generation: the native forward wins the
# expansion 1x1s; wrw v1 and the dilated stride-2 gradin lose (kernel
# work tracked in docs/ROADMAP; re-measure + regenerate after each
# kernel change).
_CONV_TABLE: dict = {
    (64, 64, 1, 1): {"fwd": False, "gradin": False, "wrw": False},
    (64, 64, 3, 1): {"fwd": False, "gradin": False, "wrw": True},
    (64, 256, 1, 1): {"fwd": True, "gradin": False, "wrw": False},
    (128, 128, 3, 1): {"fwd": False, "gradin": False, "wrw": False},
    (128, 128, 3, 2): {"fwd": True, "gradin": False, "wrw": False},
    (128, 512, 1, 1): {"fwd": True, "gradin": False, "wrw": False},
    (256, 64, 1, 1): {"fwd": False, "gradin": True, "wrw": False},
    (256, 128, 1, 1): {"fwd": True, "gradin": False, "wrw": False},
    (256, 256, 3, 1): {"fwd": False, "gradin": False, "wrw": False},
    (256, 256, 3, 2): {"fwd": False, "gradin": False, "wrw": False},
    (256, 512, 1, 2): {"fwd": True, "gradin": False, "wrw": False},
    (256, 1024, 1, 1): {"fwd": False, "gradin": False, "wrw": False},
    (512, 128, 1, 1): {"fwd": False, "gradin": False, "wrw": False},
    (512, 256, 1, 1): {"fwd": True, "gradin": False, "wrw": False},
    (512, 512, 3, 1): {"fwd": False, "gradin": False, "wrw": False},
    (512, 512, 3, 2): {"fwd": False, "gradin": False, "wrw": False},
    (512, 1024, 1, 2): {"fwd": True, "gradin": False, "wrw": False},
    (512, 2048, 1, 1): {"fwd": False, "gradin": False, "wrw": False},
    (1024, 256, 1, 1): {"fwd": False, "gradin": False, "wrw": False},
    (1024, 512, 1, 1): {"fwd": False, "gradin": False, "wrw": False},
    (1024, 2048, 1, 2): {"fwd": False, "gradin": False, "wrw": False},
    (2048, 512, 1, 1): {"fwd": False, "gradin": False, "wrw": False},
}
_load_table()


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
        need_gx = ctx.needs_input_grad[0]
        need_gw = ctx.needs_input_grad[1]
        gx = gw = gb = None
        fused = None
        if hasattr(backend, "both") and (need_gx or need_gw):
            fused = backend.both(gy, x, w_c, stride, pad, need_gx,
                                 need_gw)
        if fused is not None:
            gx, gw = fused
        else:
            if need_gx:
                gx = backend.gradin(gy, w_c, (x.shape[2], x.shape[3]),
                                    stride, pad)
            if need_gw:
                gw = backend.wrw(gy, x, w_c.shape, stride, pad)
        if gw is not None:
            gw = TF.grad_mask_apply(gw, mask, ctx.weight_dtype)
        if ctx.bias_dtype is not None and ctx.needs_input_grad[3]:
            gb = gy.float().sum(dim=(0, 2, 3)).to(ctx.bias_dtype)
        return gx, gw, None, gb, None, None, None, None


def masked_conv2d_native(x, weight, mask, bias, stride: int, pad: int,
                         w_cache: torch.Tensor,
                         backend=None) -> torch.Tensor:
    if backend is None:
        backend = (NativeBackend
                   if os.environ.get("TURBOPRUNE_CONV", "auto") == "native"
                   else AutoBackend)
    return MaskedConvNativeFn.apply(x, weight, mask, bias, stride, pad,
                                    w_cache, backend)


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
    """Full gate for routing ConvMask through MaskedConvNativeFn.

    TURBOPRUNE_CONV modes: ``auto`` (default — the measured per-op
    dispatch table, MIOpen where it wins), ``native`` (all three ops on
    the in-house kernels, A/B knob), ``off``/``library`` (plain library
    conv path). Plus bf16 channels_last GPU tensors + shape_ok."""
    mode = os.environ.get("TURBOPRUNE_CONV", "auto")
    if mode not in ("auto", "native"):
        return False
    if not (x.is_cuda and x.dtype == torch.bfloat16
            and w.dtype == torch.bfloat16):
        return False
    if not shape_ok(w.shape[0], w.shape[1], w.shape[2], w.shape[3],
                    stride, padding, dilation, groups):
        return False
    return x.is_contiguous(memory_format=torch.channels_last)
