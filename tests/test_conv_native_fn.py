"""ConvImplicitFn wiring oracle on CPU: with the TorchBackend the whole
autograd Function (fwd + grad_input composition + wrw + bias grad) must
reproduce F.conv2d autograd exactly in fp32. The GPU path swaps only the
backend, so this pins every piece of the wiring the kernels plug into."""

import pytest
import torch
import torch.nn.functional as F

from turboprune_amd.ops import conv_native

GEOMS = [  # (Cin, Cout, k, stride, bias)
    (8, 16, 3, 1, False),
    (8, 16, 3, 2, True),
    (16, 8, 1, 1, True),
    (16, 8, 1, 2, False),
]


@pytest.mark.parametrize("cin,cout,k,stride,bias", GEOMS)
def test_fn_matches_autograd(cin, cout, k, stride, bias):
    torch.manual_seed(0)
    pad = k // 2
    x = torch.randn(2, cin, 14, 14, requires_grad=True)
    w = torch.randn(cout, cin, k, k, requires_grad=True)
    b = torch.randn(cout, requires_grad=True) if bias else None

    y = conv_native.conv2d(x, w, b, stride, pad)
    y_ref = F.conv2d(x, w, b, stride, pad)
    torch.testing.assert_close(y, y_ref)

    gy = torch.randn_like(y)
    grads = torch.autograd.grad(y, [x, w] + ([b] if bias else []), gy)
    refs = torch.autograd.grad(y_ref, [x, w] + ([b] if bias else []), gy)
    for g, r in zip(grads, refs):
        torch.testing.assert_close(g, r, rtol=1e-4, atol=1e-4)


def test_no_input_grad_needed():
    # first-layer case: x is a leaf without requires_grad
    x = torch.randn(2, 8, 8, 8)
    w = torch.randn(16, 8, 3, 3, requires_grad=True)
    y = conv_native.conv2d(x, w, None, 1, 1)
    y.sum().backward()
    assert w.grad is not None


def test_native_dispatch_is_gpu_gated(monkeypatch):
    monkeypatch.setenv("TURBOPRUNE_CONV", "native")
    x = torch.randn(2, 64, 8, 8)
    w = torch.randn(64, 64, 3, 3)
    # CPU fp32 tensors: guard must refuse even with the flag set
    assert not conv_native.native_conv_ok(
        x, w, (1, 1), (1, 1), (1, 1), 1)


def test_masked_convmask_through_fn_cpu_backend():
    """End-to-end: masked weight chain -> ConvImplicitFn; grad_weight
    must still be mask * dense-grad (the MaskedWeight contract)."""
    from turboprune_amd.ops import functional as TF
    from turboprune_amd.ops.mask_layers import ConvMask
    torch.manual_seed(1)
    layer = ConvMask(in_channels=8, out_channels=16, kernel_size=3,
                     padding=1, bias=False)
    layer.mask.bernoulli_(0.5)
    x = torch.randn(2, 8, 10, 10)
    w = TF.masked_weight(layer.weight, layer.mask, None, None)
    y = conv_native.conv2d(x, w, None, 1, 1)
    y.sum().backward()
    assert torch.all(layer.weight.grad[layer.mask == 0] == 0)
    w2 = layer.weight.detach().clone().requires_grad_(True)
    F.conv2d(x, layer.mask * w2, None, 1, 1).sum().backward()
    torch.testing.assert_close(layer.weight.grad, w2.grad * layer.mask,
                               rtol=1e-4, atol=1e-4)


def test_resnet50_dispatch_envelope():
    """Exactly the stem (Cin=3) falls outside the native-conv shape
    envelope on ResNet50; every other conv routes to the MFMA triple
    once the GPU/bf16 gate opens."""
    import torch.nn as nn

    from turboprune_amd.config import compose
    from turboprune_amd.models import build_model
    from turboprune_amd.ops.conv_native import shape_ok
    pm = build_model(compose("bench_resnet50_imagenet"))
    inside, outside = [], []
    for name, m in pm.model.named_modules():
        if isinstance(m, nn.Conv2d):
            ok = shape_ok(m.out_channels, m.in_channels,
                          m.kernel_size[0], m.kernel_size[1], m.stride,
                          m.padding, m.dilation, m.groups)
            (inside if ok else outside).append(name)
    assert outside == ["conv1"], outside
    assert len(inside) == 52, len(inside)


def test_auto_backend_table_logic():
    """Dispatch-table plumbing (no GPU needed): measured winners route
    per (shape, op); unknown shapes use the declared defaults; the
    Function is skipped entirely for all-library shapes."""
    from turboprune_amd.ops.conv_native import AutoBackend, _load_table
    _load_table()
    t = AutoBackend.table
    assert len(t) >= 20
    # measured winners (profiles/r02_conv_dispatch.md + r2r)
    assert t[(64, 256, 1, 1)]["fwd"] is True
    assert t[(64, 64, 3, 1)]["wrw"] is True      # tr_b16 v6 beat MIOpen
    assert t[(64, 64, 3, 1)]["fwd"] is False
    assert t[(512, 512, 3, 1)]["wrw"] is False   # MIOpen keeps it
    # any_native gates the Function entry
    assert AutoBackend.any_native(256, 64, 1, 1)       # fwd native
    assert AutoBackend.any_native(64, 64, 3, 1)        # wrw native
    assert not AutoBackend.any_native(64, 64, 1, 1)    # all library
    # unknown shape: defaults
    d = AutoBackend._ops_for((9999, 9999, 3, 1))
    assert d == AutoBackend.DEFAULT_NATIVE


def test_auto_table_json_override(tmp_path, monkeypatch):
    import json
    p = tmp_path / "table.json"
    p.write_text(json.dumps({"[1, 2, 3, 4]": {"fwd": True,
                                              "gradin": False,
                                              "wrw": False}}))
    monkeypatch.setenv("TURBOPRUNE_CONV_TABLE", str(p))
    from turboprune_amd.ops.conv_native import AutoBackend, _load_table
    try:
        _load_table()
        assert AutoBackend.table[(1, 2, 3, 4)]["fwd"] is True
    finally:
        monkeypatch.delenv("TURBOPRUNE_CONV_TABLE")
        _load_table()  # restore the built-in table
