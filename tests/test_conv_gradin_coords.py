"""Functional CPU check of conv2d_implicit_gradin's coordinate logic
(csrc/conv_implicit.hip DIL>1 path, written off-device): replicate the
kernel's EXACT per-element source selection — dilated-space bounds test,
divisibility test, compact-index division, rotated-transposed weight,
new_pad — as an explicit gather + matmul, and compare the resulting
grad_input against autograd. The LDS/MFMA machinery around this math is
shared with the device-validated forward kernel; the coordinates are
what this file pins."""

import pytest
import torch
import torch.nn.functional as F


def gradin_via_kernel_coords(gy, w, Hi, Wi, stride, pad):
    """Mirror of the kernel: output pixel opix=(n,hi_out,wi_out) over
    (N,Hi,Wi), contraction over (dh,dw,cout); source = gy[n, hd/DIL,
    wd/DIL] iff dilated coords in bounds and divisible, else 0."""
    N, Cout, Ho, Wo = gy.shape
    Cin, _, KH, KW = w.shape[1], None, w.shape[2], w.shape[3]
    k = KH
    new_pad = k - 1 - pad
    Hd = Hi + 2 * pad - k + 1
    Wd = Wi + 2 * pad - k + 1
    # rotated-transposed weight as the kernel's B operand:
    # (Cin, KH*KW*Cout) tap-major (channels_last memory of w_rt)
    w_rt = torch.flip(w, dims=[2, 3]).permute(1, 0, 2, 3)  # (Cin,Cout,k,k)
    B = w_rt.permute(0, 2, 3, 1).reshape(Cin, k * k * Cout)

    gx = torch.zeros(N, Cin, Hi, Wi)
    for n in range(N):
        for ho in range(Hi):          # output pixel rows = input rows
            for wo in range(Wi):
                a = torch.zeros(k * k * Cout)
                for dh in range(k):
                    for dw in range(k):
                        hi = ho * 1 - new_pad + dh   # kernel: stride=1
                        wi = wo * 1 - new_pad + dw
                        ok = 0 <= hi < Hd and 0 <= wi < Wd
                        if ok and stride > 1:
                            if hi % stride or wi % stride:
                                ok = False
                            else:
                                hi //= stride
                                wi //= stride
                        if ok:
                            tap = dh * k + dw
                            a[tap * Cout:(tap + 1) * Cout] = gy[n, :, hi, wi]
                gx[n, :, ho, wo] = B @ a
    return gx


@pytest.mark.parametrize("k,stride,pad,hi", [
    (3, 1, 1, 8), (3, 2, 1, 9), (1, 1, 0, 6), (1, 2, 0, 8),
])
def test_kernel_coordinate_math_matches_autograd(k, stride, pad, hi):
    torch.manual_seed(0)
    cin, cout = 3, 4
    x = torch.randn(2, cin, hi, hi, requires_grad=True)
    w = torch.randn(cout, cin, k, k)
    y = F.conv2d(x, w, None, stride, pad)
    gy = torch.randn_like(y)
    y.backward(gy)
    gx = gradin_via_kernel_coords(gy, w, hi, hi, stride, pad)
    torch.testing.assert_close(gx, x.grad, rtol=1e-4, atol=1e-4)


def conv_fwd_via_kernel_coords(x, w, stride, pad):
    """Mirror of conv3x3_fwd_kernel's im2col gather (DIL=1): source =
    x[n, ho*stride-pad+dh, wo*stride-pad+dw] with zero-page fallback;
    B = channels_last weight viewed (Cout, KH*KW*Cin) tap-major."""
    N, Cin, Hi, Wi = x.shape
    Cout, _, KH, KW = w.shape
    Ho = (Hi + 2 * pad - KH) // stride + 1
    Wo = (Wi + 2 * pad - KW) // stride + 1
    B = w.permute(0, 2, 3, 1).reshape(Cout, KH * KW * Cin)
    y = torch.zeros(N, Cout, Ho, Wo)
    for n in range(N):
        for ho in range(Ho):
            for wo in range(Wo):
                a = torch.zeros(KH * KW * Cin)
                for dh in range(KH):
                    for dw in range(KW):
                        hi = ho * stride - pad + dh
                        wi = wo * stride - pad + dw
                        if 0 <= hi < Hi and 0 <= wi < Wi:
                            tap = dh * KW + dw
                            a[tap * Cin:(tap + 1) * Cin] = x[n, :, hi, wi]
                y[n, :, ho, wo] = B @ a
    return y


@pytest.mark.parametrize("k,stride,pad", [(3, 1, 1), (3, 2, 1),
                                          (1, 1, 0), (1, 2, 0),
                                          (7, 2, 3)])
def test_fwd_kernel_coordinate_math(k, stride, pad):
    torch.manual_seed(1)
    x = torch.randn(2, 3, 12, 12)
    w = torch.randn(4, 3, k, k)
    got = conv_fwd_via_kernel_coords(x, w, stride, pad)
    ref = F.conv2d(x, w, None, stride, pad)
    torch.testing.assert_close(got, ref, rtol=1e-4, atol=1e-4)
