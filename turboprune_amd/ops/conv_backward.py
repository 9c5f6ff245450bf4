"""Conv backward as forward-conv compositions (implicit-GEMM plumbing).

grad_input of conv(x, w, stride, pad) equals a FORWARD convolution of
the (zero-dilated, for stride > 1) output gradient with the
rotated-and-transposed kernel:

    gx = conv( dilate_s(gy) zero-padded to input geometry,
               rot180(w)^T, stride=1, pad=k-1-pad )

With stride 1 the dilation is a no-op, so the GPU path is exactly the
validated implicit-GEMM forward kernel with a permuted weight
(scripts/validate_conv_implicit.py measured it faster than MIOpen's
conv-backward-input on 3 of 5 ResNet50 shapes). This module carries the
backend-agnostic math; pass ``conv_fn=F.conv2d`` (the CPU oracle used in
tests) or a lambda over ``_C.conv2d_implicit_fwd``.
"""

from __future__ import annotations

from typing import Callable, Optional, Tuple

import torch
import torch.nn.functional as F


def rot180_transpose(w: torch.Tensor) -> torch.Tensor:
    """(Cout, Cin, KH, KW) -> (Cin, Cout, KH, KW) with spatially flipped
    taps; channels_last is preserved when the input is channels_last."""
    out = torch.flip(w, dims=[2, 3]).permute(1, 0, 2, 3)
    if w.is_contiguous(memory_format=torch.channels_last):
        return out.contiguous(memory_format=torch.channels_last)
    return out.contiguous()


def dilate_gy(gy: torch.Tensor, stride: int,
              in_hw: Tuple[int, int], k: int, pad: int) -> torch.Tensor:
    """Insert (stride-1) zeros between output-gradient pixels and pad the
    tail so a stride-1 conv with pad k-1-pad reproduces the input
    geometry: dilated size must be Hi + 2*pad - k + 1."""
    if stride == 1:
        return gy
    n, c, ho, wo = gy.shape
    hi, wi = in_hw
    hd = hi + 2 * pad - k + 1
    wd = wi + 2 * pad - k + 1
    out = gy.new_zeros(n, c, hd, wd)
    out[:, :, : (ho - 1) * stride + 1 : stride,
        : (wo - 1) * stride + 1 : stride] = gy
    if gy.is_contiguous(memory_format=torch.channels_last):
        out = out.contiguous(memory_format=torch.channels_last)
    return out


def conv_grad_input(gy: torch.Tensor, w: torch.Tensor,
                    in_hw: Tuple[int, int], stride: int, pad: int,
                    conv_fn: Optional[Callable] = None) -> torch.Tensor:
    """grad wrt the conv input, any stride, via a forward conv."""
    conv_fn = conv_fn or (lambda a, b, s, p: F.conv2d(a, b, None, s, p))
    k = w.shape[2]
    w_rt = rot180_transpose(w)
    gy_d = dilate_gy(gy, stride, in_hw, k, pad)
    return conv_fn(gy_d, w_rt, 1, k - 1 - pad)
