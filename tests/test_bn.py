"""FusedBatchNorm2d / bn_act: CPU composed-path semantics vs
nn.BatchNorm2d, and (GPU) the fused HIP kernels vs the fp32 torch
oracle."""

import copy

import pytest
import torch
import torch.nn as nn

from turboprune_amd.ops.bn import FusedBatchNorm2d, bn_act


def test_composed_matches_module_cpu():
    torch.manual_seed(0)
    ours = FusedBatchNorm2d(8)
    ref = nn.BatchNorm2d(8)
    ref.load_state_dict(ours.state_dict())
    for step in range(3):
        x = torch.randn(4, 8, 6, 6)
        y1 = ours(x)
        y2 = ref(x)
        assert torch.allclose(y1, y2, atol=1e-6)
    assert torch.allclose(ours.running_mean, ref.running_mean)
    assert torch.allclose(ours.running_var, ref.running_var)
    assert ours.num_batches_tracked == ref.num_batches_tracked
    ours.eval(); ref.eval()
    x = torch.randn(4, 8, 6, 6)
    assert torch.allclose(ours(x), ref(x), atol=1e-6)


def test_bn_act_relu_residual_cpu():
    torch.manual_seed(1)
    bn = FusedBatchNorm2d(4)
    x = torch.randn(2, 4, 5, 5, requires_grad=True)
    res = torch.randn(2, 4, 5, 5, requires_grad=True)
    y = bn_act(bn, x, residual=res, relu=True)
    ref_bn = nn.BatchNorm2d(4)
    ref_bn.load_state_dict({k: v for k, v in bn.state_dict().items()})
    # rebuild reference forward from scratch stats (fresh running stats)
    ref_bn.running_mean.zero_(); ref_bn.running_var.fill_(1.0)
    ref_bn.num_batches_tracked.zero_()
    x2 = x.detach().clone().requires_grad_()
    r2 = res.detach().clone().requires_grad_()
    y_ref = torch.relu(ref_bn(x2) + r2)
    assert torch.allclose(y, y_ref, atol=1e-6)
    y.sum().backward()
    y_ref.sum().backward()
    assert torch.allclose(x.grad, x2.grad, atol=1e-6)
    assert torch.allclose(res.grad, r2.grad, atol=1e-6)


@pytest.mark.gpu
@pytest.mark.parametrize("relu", [False, True])
@pytest.mark.parametrize("with_res", [False, True])
@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_fused_bn_matches_oracle_gpu(relu, with_res, dtype):
    torch.manual_seed(0)
    dev = "cuda:0"
    C = 96
    bn = FusedBatchNorm2d(C).to(dev)
    ref = nn.BatchNorm2d(C).to(dev)
    ref.load_state_dict(bn.state_dict())

    x = torch.randn(8, C, 14, 14, device=dev).to(dtype) \
        .to(memory_format=torch.channels_last).requires_grad_()
    res = None
    x2 = x.detach().clone().requires_grad_()
    r2 = None
    if with_res:
        res = torch.randn(8, C, 14, 14, device=dev).to(dtype) \
            .to(memory_format=torch.channels_last).requires_grad_()
        r2 = res.detach().clone().requires_grad_()

    y = bn_act(bn, x, residual=res, relu=relu)
    y_ref = ref(x2.float())
    if with_res:
        y_ref = y_ref + r2.float()
    if relu:
        y_ref = torch.relu(y_ref)

    atol = 1e-5 if dtype == torch.float32 else 5e-2
    assert (y.float() - y_ref).abs().max().item() < atol
    assert torch.allclose(bn.running_mean, ref.running_mean, atol=1e-3)
    assert torch.allclose(bn.running_var, ref.running_var, atol=1e-3)

    # feed BOTH sides the same (dtype-rounded) dy so dgamma/dbeta sums
    # are comparable
    dy = torch.randn_like(y_ref).to(dtype)
    y.backward(dy)
    y_ref.backward(dy.float())
    gtol = 1e-4 if dtype == torch.float32 else 8e-2
    assert (x.grad.float() - x2.grad).abs().max().item() < gtol
    if with_res:
        assert (res.grad.float() - r2.grad).abs().max().item() < gtol
    assert torch.allclose(bn.weight.grad, ref.weight.grad,
                          atol=1e-2 if dtype == torch.bfloat16 else 1e-3,
                          rtol=1e-2)
    assert torch.allclose(bn.bias.grad, ref.bias.grad,
                          atol=1e-2 if dtype == torch.bfloat16 else 1e-3,
                          rtol=1e-2)


@pytest.mark.gpu
def test_fused_bn_eval_mode_gpu():
    torch.manual_seed(2)
    dev = "cuda:0"
    bn = FusedBatchNorm2d(32).to(dev)
    ref = nn.BatchNorm2d(32).to(dev)
    # give nontrivial running stats
    with torch.no_grad():
        bn.running_mean.normal_(); bn.running_var.uniform_(0.5, 2.0)
    ref.load_state_dict(bn.state_dict())
    bn.eval(); ref.eval()
    x = torch.randn(4, 32, 8, 8, device=dev) \
        .to(memory_format=torch.channels_last)
    with torch.no_grad():
        y = bn_act(bn, x, relu=True)
        y_ref = torch.relu(ref(x))
    assert (y - y_ref).abs().max().item() < 1e-4


@pytest.mark.gpu
def test_resnet_block_uses_fused_bn_gpu():
    """A ResNet50 training step on channels_last bf16 runs the fused BN
    path and produces finite grads."""
    from turboprune_amd.models import resnet50
    m = resnet50(num_classes=10).to("cuda:0") \
        .to(memory_format=torch.channels_last)
    x = torch.randn(4, 3, 64, 64, device="cuda:0") \
        .to(memory_format=torch.channels_last)
    with torch.autocast(device_type="cuda", dtype=torch.bfloat16):
        out = m(x)
    out.float().sum().backward()
    torch.cuda.synchronize()
    assert all(torch.isfinite(p.grad).all() for p in m.parameters()
               if p.grad is not None)
    assert int(m.bn1.num_batches_tracked) == 1
