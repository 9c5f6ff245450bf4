"""FusedMaskedSGD must match torch.optim.SGD bit-for-bit on CPU, and the
ScheduleFree optimizer must converge + swap iterates correctly."""

import copy

import torch
import torch.nn as nn

from turboprune_amd.ops.mask_layers import LinearMask
from turboprune_amd.optim import FusedMaskedSGD, ScheduleFreeSGD


def _train_steps(model, opt, steps=5, seed=0):
    torch.manual_seed(seed)
    for _ in range(steps):
        x = torch.randn(8, 10)
        loss = model(x).pow(2).mean()
        opt.zero_grad()
        loss.backward()
        opt.step()


def test_fused_sgd_matches_torch_sgd():
    torch.manual_seed(0)
    m1 = nn.Sequential(nn.Linear(10, 16), nn.ReLU(), nn.Linear(16, 4))
    m2 = copy.deepcopy(m1)
    o1 = torch.optim.SGD(m1.parameters(), lr=0.1, momentum=0.9,
                         weight_decay=5e-4)
    o2 = FusedMaskedSGD(m2.parameters(), lr=0.1, momentum=0.9,
                        weight_decay=5e-4)
    _train_steps(m1, o1, seed=1)
    _train_steps(m2, o2, seed=1)
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p1, p2, atol=1e-7), (p1 - p2).abs().max()


def test_fused_sgd_keeps_cache_valid():
    torch.manual_seed(0)
    model = nn.Sequential(LinearMask(in_features=10, out_features=4))
    layer = model[0]
    layer.mask.bernoulli_(0.5)
    layer.enable_cache(torch.float32)
    opt = FusedMaskedSGD(model.parameters(), lr=0.1, momentum=0.9,
                         weight_decay=0.0, model=model)
    _train_steps(model, opt, steps=3)
    expected = layer.mask * layer.weight
    assert torch.allclose(layer.weight_masked, expected)


def test_fused_sgd_state_dict_roundtrip_with_torch_sgd():
    model = nn.Linear(10, 4)
    opt = FusedMaskedSGD(model.parameters(), lr=0.1, momentum=0.9)
    _train_steps(model, opt, steps=2, seed=3)
    sd = opt.state_dict()
    # loads into a plain torch SGD (artifact format compatibility)
    opt2 = torch.optim.SGD(model.parameters(), lr=0.1, momentum=0.9)
    opt2.load_state_dict(sd)
    opt3 = FusedMaskedSGD(model.parameters(), lr=0.1, momentum=0.9)
    opt3.load_state_dict(opt2.state_dict())


def test_masked_weights_keep_receiving_updates():
    """Reference semantics: masked-out weights still get wd/momentum
    updates (nullified only at forward)."""
    torch.manual_seed(0)
    model = nn.Sequential(LinearMask(in_features=6, out_features=3))
    layer = model[0]
    layer.mask.zero_()  # fully masked
    w_before = layer.weight.detach().clone()
    opt = FusedMaskedSGD(model.parameters(), lr=0.1, momentum=0.0,
                         weight_decay=0.1, model=model)
    x = torch.randn(4, 6)
    loss = model(x).pow(2).mean()
    opt.zero_grad()
    loss.backward()
    opt.step()
    # forward grad is zero (mask=0) but weight decay still shrinks weights
    assert not torch.allclose(layer.weight, w_before)
    assert torch.allclose(layer.weight, w_before * (1 - 0.1 * 0.1))


def test_schedule_free_train_eval_swap():
    torch.manual_seed(0)
    model = nn.Linear(5, 1)
    opt = ScheduleFreeSGD(model.parameters(), lr=0.05, momentum=0.9)
    opt.train()
    for _ in range(10):
        x = torch.randn(16, 5)
        loss = (model(x) - 1).pow(2).mean()
        opt.zero_grad()
        loss.backward()
        opt.step()
    w_train = model.weight.detach().clone()
    opt.eval()
    w_eval = model.weight.detach().clone()
    assert not torch.allclose(w_train, w_eval)
    opt.train()
    assert torch.allclose(model.weight, w_train, atol=1e-5)


def test_schedule_free_decreases_loss():
    torch.manual_seed(0)
    model = nn.Linear(3, 1)
    opt = ScheduleFreeSGD(model.parameters(), lr=0.1, momentum=0.9)
    opt.train()
    x = torch.randn(64, 3)
    y = x @ torch.tensor([[1.0], [2.0], [-1.0]])
    first = None
    for i in range(50):
        loss = (model(x) - y).pow(2).mean()
        opt.zero_grad()
        loss.backward()
        opt.step()
        if first is None:
            first = loss.item()
    assert loss.item() < first * 0.2


def test_multi_sgd_env_flag_cpu_fallback(monkeypatch):
    """TURBOPRUNE_MULTI_SGD=1 must not change CPU behavior (the multi
    path is GPU-only); parity vs torch.optim.SGD still holds."""
    import torch

    from turboprune_amd.ops.mask_layers import LinearMask
    from turboprune_amd.optim import FusedMaskedSGD
    monkeypatch.setenv("TURBOPRUNE_MULTI_SGD", "1")
    torch.manual_seed(11)
    m = LinearMask(in_features=6, out_features=4, bias=True)
    m.mask.bernoulli_(0.5)
    ref = torch.nn.Linear(6, 4)
    ref.load_state_dict({"weight": m.weight.detach().clone(),
                         "bias": m.bias.detach().clone()})
    opt = FusedMaskedSGD(m.parameters(), lr=0.1, momentum=0.9,
                         weight_decay=1e-4, model=m)
    ropt = torch.optim.SGD(ref.parameters(), lr=0.1, momentum=0.9,
                           weight_decay=1e-4)
    for _ in range(3):
        g = torch.randn_like(m.weight)
        m.weight.grad = g.clone()
        m.bias.grad = torch.randn_like(m.bias)
        ref.weight.grad = g.clone()
        ref.bias.grad = m.bias.grad.clone()
        opt.step()
        ropt.step()
    assert torch.allclose(m.weight, ref.weight, atol=1e-6)


def test_multi_ok_layout_guard():
    import torch

    from turboprune_amd.optim.sgd import FusedMaskedSGD
    p = torch.randn(4, 3, 3, 3)
    good = torch.randn_like(p)
    bad = torch.randn(4, 3, 3, 3).to(memory_format=torch.channels_last)
    assert FusedMaskedSGD._multi_ok(p, good, None, None, None)
    assert not FusedMaskedSGD._multi_ok(p, bad, None, None, None)
    assert not FusedMaskedSGD._multi_ok(p.to(torch.bfloat16), good, None,
                                        None, None)
