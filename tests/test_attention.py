"""Attention wiring oracles (CPU): the closed-form recompute backward
must equal autograd through explicit softmax attention, and the sdpa()
helper must be exactly torch SDPA when the native path is off."""

import math

import pytest
import torch
import torch.nn.functional as F

from turboprune_amd.ops import attention


@pytest.mark.parametrize("s", [8, 64, 197])
def test_attn_backward_matches_autograd(s):
    torch.manual_seed(0)
    B, H, D = 2, 3, 64
    q = torch.randn(B, H, s, D, requires_grad=True)
    k = torch.randn(B, H, s, D, requires_grad=True)
    v = torch.randn(B, H, s, D, requires_grad=True)
    scale = 1.0 / math.sqrt(D)
    out = torch.softmax((q @ k.transpose(-2, -1)) * scale, dim=-1) @ v
    do = torch.randn_like(out)
    out.backward(do)

    gq, gk, gv = attention.attn_backward(
        q.detach(), k.detach(), v.detach(), do, scale)
    torch.testing.assert_close(gq, q.grad, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(gk, k.grad, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(gv, v.grad, rtol=1e-4, atol=1e-4)


def test_sdpa_default_is_torch(monkeypatch):
    monkeypatch.delenv("TURBOPRUNE_ATTN", raising=False)
    torch.manual_seed(1)
    q = torch.randn(2, 3, 17, 64)
    k = torch.randn(2, 3, 17, 64)
    v = torch.randn(2, 3, 17, 64)
    torch.testing.assert_close(
        attention.sdpa(q, k, v), F.scaled_dot_product_attention(q, k, v))


def test_native_gate_refuses_cpu(monkeypatch):
    monkeypatch.setenv("TURBOPRUNE_ATTN", "native")
    q = torch.randn(1, 1, 8, 64)
    assert not attention._native_ok(q)          # CPU fp32
    assert not attention._native_ok(q.to(torch.bfloat16))  # CPU bf16
