"""Masked layer numerics vs the reference semantics oracle:
forward must equal F.conv2d/F.linear(x, mask*weight) exactly in fp32,
and grad_weight must equal mask * dense-grad (reference:
utils/mask_layers.py:25-34,59-70,104-119)."""

import torch
import torch.nn.functional as F

from turboprune_amd.ops.mask_layers import Conv1dMask, ConvMask, LinearMask


def _random_mask(t, p=0.5):
    return torch.zeros_like(t).bernoulli_(p)


def test_convmask_forward_matches_oracle():
    torch.manual_seed(0)
    layer = ConvMask(in_channels=3, out_channels=8, kernel_size=3,
                     padding=1, bias=False)
    layer.mask.copy_(_random_mask(layer.weight))
    x = torch.randn(4, 3, 16, 16)
    y = layer(x)
    y_ref = F.conv2d(x, layer.mask * layer.weight, None, 1, 1)
    assert torch.equal(y, y_ref)


def test_convmask_backward_masks_gradient():
    torch.manual_seed(1)
    layer = ConvMask(in_channels=3, out_channels=4, kernel_size=3, bias=False)
    layer.mask.copy_(_random_mask(layer.weight))
    x = torch.randn(2, 3, 8, 8, requires_grad=True)
    layer(x).sum().backward()
    # gradient is zero exactly where the mask is zero
    assert torch.all(layer.weight.grad[layer.mask == 0] == 0)

    # and equals the dense gradient times the mask
    w = layer.weight.detach().clone().requires_grad_(True)
    y = F.conv2d(x.detach(), layer.mask * w)
    y.sum().backward()
    assert torch.allclose(layer.weight.grad, w.grad * layer.mask)


def test_linearmask_forward_backward():
    torch.manual_seed(2)
    layer = LinearMask(in_features=10, out_features=7, bias=True)
    layer.mask.copy_(_random_mask(layer.weight))
    x = torch.randn(5, 10)
    y = layer(x)
    assert torch.equal(y, F.linear(x, layer.mask * layer.weight, layer.bias))
    y.sum().backward()
    assert torch.all(layer.weight.grad[layer.mask == 0] == 0)


def test_conv1dmask_linear_equivalence_2d_and_3d():
    torch.manual_seed(3)
    layer = Conv1dMask(in_features=12, out_features=6, bias=True)
    layer.mask.copy_(_random_mask(layer.weight))
    assert layer.weight.shape == (6, 12, 1)  # reference weight shape
    x2 = torch.randn(4, 12)
    y2 = layer(x2)
    y2_ref = F.conv1d(x2.unsqueeze(-1),
                      layer.mask * layer.weight, layer.bias).squeeze(-1)
    assert torch.allclose(y2, y2_ref, atol=1e-6)
    # (B, N, C) token input — the fixed DeiT path
    x3 = torch.randn(2, 5, 12)
    y3 = layer(x3)
    assert y3.shape == (2, 5, 6)
    y3_ref = F.linear(x3, (layer.mask * layer.weight).squeeze(-1), layer.bias)
    assert torch.allclose(y3, y3_ref, atol=1e-6)


def test_set_er_mask_density():
    torch.manual_seed(4)
    layer = ConvMask(in_channels=64, out_channels=64, kernel_size=3,
                     bias=False)
    layer.set_er_mask(0.3)
    density = layer.mask.mean().item()
    assert abs(density - 0.3) < 0.02
    assert set(layer.mask.unique().tolist()) <= {0.0, 1.0}


def test_mask_rides_in_state_dict_as_fp32():
    layer = ConvMask(in_channels=2, out_channels=2, kernel_size=1)
    sd = layer.state_dict()
    assert "mask" in sd
    assert sd["mask"].dtype == torch.float32
    assert sd["mask"].shape == layer.weight.shape
    assert "weight_masked" not in sd  # cache never serialized


def test_cache_matches_uncached_forward():
    torch.manual_seed(5)
    layer = ConvMask(in_channels=3, out_channels=4, kernel_size=3, bias=False)
    layer.mask.copy_(_random_mask(layer.weight))
    x = torch.randn(2, 3, 8, 8)
    y_ref = layer(x)
    layer.enable_cache(torch.float32)
    y_cached = layer(x)
    assert torch.equal(y_cached, y_ref)
    # cache survives a mask rewrite via refresh
    layer.mask.zero_()
    layer.refresh_cache()
    assert torch.all(layer(x) == 0)


def test_cache_backward_still_masks_grad():
    torch.manual_seed(6)
    layer = LinearMask(in_features=8, out_features=3, bias=False)
    layer.mask.copy_(_random_mask(layer.weight))
    layer.enable_cache(torch.float32)
    x = torch.randn(4, 8)
    layer(x).sum().backward()
    assert layer.weight.grad is not None
    assert torch.all(layer.weight.grad[layer.mask == 0] == 0)


def test_sparsity_accounting():
    layer = ConvMask(in_channels=4, out_channels=4, kernel_size=3, bias=False)
    layer.mask.zero_()
    layer.mask.view(-1)[: layer.mask.numel() // 4].fill_(1)
    assert abs(layer.sparsity() - 0.75) < 1e-6


def test_cache_staleness_guard_on_foreign_mutation():
    """A dispatcher-level weight mutation (foreign optimizer, manual
    add_) must invalidate the masked-weight cache automatically."""
    torch.manual_seed(9)
    layer = LinearMask(in_features=6, out_features=3, bias=False)
    layer.mask.bernoulli_(0.5)
    layer.enable_cache(torch.float32)
    x = torch.randn(2, 6)
    y0 = layer(x)
    with torch.no_grad():
        layer.weight.add_(1.0)  # bumps _version; cache NOT refreshed here
    y1 = layer(x)
    ref = torch.nn.functional.linear(x, layer.mask * layer.weight)
    assert torch.allclose(y1, ref)
    assert not torch.allclose(y0, y1)
