"""conv_backward math oracle: grad_input via forward-conv composition
(rot180-transposed weight + zero-dilated gy) must equal autograd's
grad for every ResNet50 conv geometry. CPU uses F.conv2d as the conv
backend; on GPU the same composition runs over the implicit-GEMM
forward kernel (scripts/validate_conv_implicit.py)."""

import pytest
import torch
import torch.nn.functional as F

from turboprune_amd.ops.conv_backward import conv_grad_input, dilate_gy

GEOMS = [  # (Cin, Cout, k, stride, pad, Hi)
    (8, 16, 3, 1, 1, 14),
    (8, 16, 3, 2, 1, 14),
    (16, 8, 1, 1, 0, 14),
    (16, 8, 1, 2, 0, 14),
    (8, 8, 7, 2, 3, 32),   # stem
]


@pytest.mark.parametrize("cin,cout,k,stride,pad,hi", GEOMS)
def test_grad_input_matches_autograd(cin, cout, k, stride, pad, hi):
    torch.manual_seed(0)
    x = torch.randn(2, cin, hi, hi, requires_grad=True)
    w = torch.randn(cout, cin, k, k)
    y = F.conv2d(x, w, None, stride, pad)
    gy = torch.randn_like(y)
    y.backward(gy)

    gx = conv_grad_input(gy, w, (hi, hi), stride, pad)
    assert gx.shape == x.shape
    torch.testing.assert_close(gx, x.grad, rtol=1e-4, atol=1e-4)


def test_dilate_geometry_stride2():
    gy = torch.arange(2 * 3 * 4 * 4, dtype=torch.float32).view(2, 3, 4, 4)
    d = dilate_gy(gy, 2, (8, 8), 3, 1)
    # dilated size = Hi + 2*pad - k + 1 = 8
    assert d.shape == (2, 3, 8, 8)
    assert torch.equal(d[:, :, ::2, ::2][:, :, :4, :4], gy)
    assert d[:, :, 1::2].abs().sum() == 0


def test_grad_input_fuzz_geometries():
    """Randomized geometries beyond ResNet50 (k up to 5, stride up to 3,
    arbitrary pad) against autograd."""
    import random
    rng = random.Random(7)
    for _ in range(25):
        k = rng.choice([1, 3, 5])
        s = rng.choice([1, 2, 3])
        pad = rng.randint(0, k - 1) if k > 1 else 0
        cin, cout = rng.choice([4, 8]), rng.choice([4, 8])
        hi = rng.randint(k + s, 20)
        torch.manual_seed(hi)
        x = torch.randn(2, cin, hi, hi, requires_grad=True)
        w = torch.randn(cout, cin, k, k)
        y = F.conv2d(x, w, None, s, pad)
        if y.shape[-1] < 1:
            continue
        gy = torch.randn_like(y)
        y.backward(gy)
        gx = conv_grad_input(gy, w, (hi, hi), s, pad)
        torch.testing.assert_close(gx, x.grad, rtol=1e-4, atol=1e-4)


@pytest.mark.parametrize("shape", [
    # every ResNet50 stride-2 geometry (cin, cout, k, hi) at p=k//2
    (64, 128, 3, 16),
    (128, 128, 3, 56),
    (64, 256, 1, 56),
    (128, 512, 1, 28),
    (32, 64, 3, 15),   # odd spatial
])
def test_conv_grad_input_s2_parity_oracle(shape):
    """Parity-class stride-2 gradin == autograd's grad (fp32 CPU)."""
    from turboprune_amd.ops.conv_backward import conv_grad_input_s2_parity
    cin, cout, k, hi = shape
    p = k // 2
    torch.manual_seed(cin + k)
    x = torch.randn(2, cin, hi, hi, requires_grad=True)
    w = torch.randn(cout, cin, k, k, requires_grad=True) * 0.1
    y = torch.nn.functional.conv2d(x, w, None, 2, p)
    gy = torch.randn_like(y)
    (ref,) = torch.autograd.grad(y, x, gy)
    got = conv_grad_input_s2_parity(gy, w.detach(), (hi, hi), p)
    assert torch.allclose(got, ref, atol=1e-4), \
        (got - ref).abs().max().item()


def test_conv_grad_input_s2_parity_channels_last():
    from turboprune_amd.ops.conv_backward import conv_grad_input_s2_parity
    torch.manual_seed(0)
    x = torch.randn(2, 8, 12, 12, requires_grad=True)
    w = torch.randn(16, 8, 3, 3) * 0.1
    y = torch.nn.functional.conv2d(x, w, None, 2, 1)
    gy = torch.randn_like(y).contiguous(
        memory_format=torch.channels_last)
    (ref,) = torch.autograd.grad(y, x, gy)
    got = conv_grad_input_s2_parity(
        gy, w.contiguous(memory_format=torch.channels_last), (12, 12), 1)
    assert got.is_contiguous(memory_format=torch.channels_last)
    assert torch.allclose(got, ref, atol=1e-4)
