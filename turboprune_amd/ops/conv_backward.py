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


def conv_grad_input_s2_parity(gy: torch.Tensor, w: torch.Tensor,
                              in_hw: Tuple[int, int], pad: int,
                              conv_fn: Optional[Callable] = None
                              ) -> torch.Tensor:
    """Stride-2 grad_input as FOUR dense stride-1 sub-convolutions
    (output-parity classes) instead of the zero-dilated conv (which
    spends 4x the MFMA work multiplying inserted zeros — measured 3-4x
    slower than MIOpen, profiles/r02_conv_dispatch.md).

    For k=3,s=2,p=1 (and the k=1,s=2,p=0 degenerate case): an output
    pixel gx[2a+rh, 2b+rw] only receives taps with dh ≡ (rh+p) mod 2,
    so each parity class (rh, rw) is a stride-1 conv of gy with a
    (rh+1)x(rw+1) sub-kernel of the rotated-transposed weight; results
    interleave back by strided scatter. Total FLOP = the exact gradin
    FLOP, zero dilation waste."""
    conv_fn = conv_fn or (lambda a, b, s, p: F.conv2d(a, b, None, s, p))
    k = w.shape[2]
    hi, wi = in_hw
    n, cout, ho, wo = gy.shape
    cin = w.shape[1]
    cl = gy.is_contiguous(memory_format=torch.channels_last)
    wf = rot180_transpose(w)  # (Cin, Cout, k, k), taps flipped

    if k == 1:
        # only the (p, p)-parity class exists: plain 1x1 stride-1 conv
        # scattered to every stride-th position, zeros elsewhere
        assert pad == 0
        out = gy.new_zeros(n, cin, hi, wi)
        if cl:
            out = out.contiguous(memory_format=torch.channels_last)
        cls = conv_fn(gy, wf, 1, 0)
        out[:, :, ::2, ::2] = cls[:, :, :(hi + 1) // 2, :(wi + 1) // 2]
        return out

    assert k == 3 and pad == 1, "parity path covers k3p1/k1p0 stride-2"
    # flipped-tap index sets per parity (ascending gy-offset order):
    # rh=0 -> original dh=1 -> flipped index 1; rh=1 -> original {2,0}
    # -> flipped {0,2}
    sel = {0: [1], 1: [0, 2]}
    # one bottom/right-padded gy so every class runs pad-free
    gy_p = F.pad(gy, (0, 1, 0, 1))
    if cl:
        gy_p = gy_p.contiguous(memory_format=torch.channels_last)
    out = torch.empty(n, cin, hi, wi, dtype=gy.dtype, device=gy.device)
    if cl:
        out = out.contiguous(memory_format=torch.channels_last)
    for rh in (0, 1):
        for rw in (0, 1):
            sub = wf[:, :, sel[rh]][:, :, :, sel[rw]]
            sub = sub.contiguous(memory_format=torch.channels_last) \
                if cl else sub.contiguous()
            cls = conv_fn(gy_p, sub, 1, 0)
            nh = (hi - rh + 1) // 2  # rows of this parity class in gx
            nw = (wi - rw + 1) // 2
            out[:, :, rh::2, rw::2] = cls[:, :, :nh, :nw]
    return out
