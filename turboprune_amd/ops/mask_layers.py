"""Masked layers — the sparsity mechanism.

Surface-compatible with the reference (utils/mask_layers.py): classes
``ConvMask`` / ``LinearMask`` / ``Conv1dMask`` with an fp32 ``mask`` buffer
shaped like the weight (it rides in ``state_dict`` under ``...mask`` keys)
and a ``set_er_mask(p)`` Bernoulli initializer.

MI355X-first differences (observable behavior unchanged):

- Forward uses a *cached masked compute weight* when the fused optimizer
  maintains one (``weight_masked``, non-persistent, compute dtype): in
  steady state the per-forward ``mask*weight`` multiply (reference
  mask_layers.py:25 — a full weight-sized temp per forward) disappears;
  the fused SGD kernel rewrites the cache in the same pass as the update.
- Without a cache, the multiply runs as one fused HIP kernel producing the
  compute dtype directly.
- ``Conv1dMask`` keeps the reference's weight shape ``(out, in, 1)`` for
  checkpoint compatibility but evaluates as a masked linear (kernel-size-1
  conv ≡ GEMM), and also accepts (B, N, C) token inputs — fixing the
  reference's latent-broken DeiT path.

Cache contract: whoever mutates ``weight`` or ``mask`` outside the fused
optimizer must call ``refresh_cache()`` (``load_state_dict`` does this
automatically); the fused SGD step keeps the cache valid itself.
"""

from __future__ import annotations

import os
from typing import Optional

import torch
import torch.nn as nn

from turboprune_amd.ops import functional as TF


class _MaskedMixin:
    """Shared mask/cache machinery for the three masked layer types."""

    def _init_mask(self) -> None:
        self.register_buffer("mask", torch.ones_like(self.weight))
        # compute-dtype cache maintained by the fused optimizer; never saved,
        # never auto-moved (recreated on device by enable_cache/refresh_cache)
        self.weight_masked: Optional[torch.Tensor] = None
        self.compute_dtype: Optional[torch.dtype] = None
        self._w_version = -1  # staleness guard (see _fresh_cache)

    def set_er_mask(self, p: float, seed: Optional[int] = None) -> None:
        TF.bernoulli_mask_(self.mask, float(p), seed)
        self.refresh_cache()

    # --- masked-weight cache ---------------------------------------------
    def enable_cache(self, compute_dtype: torch.dtype) -> None:
        self.compute_dtype = compute_dtype
        self.weight_masked = None
        self.refresh_cache()

    def disable_cache(self) -> None:
        self.weight_masked = None
        self.compute_dtype = None

    def _fresh_cache(self) -> Optional[torch.Tensor]:
        """The cache, guaranteed fresh. The fused SGD kernel writes weight
        data in place WITHOUT bumping the tensor version (and rewrites the
        cache itself in the same pass), so an unchanged `_version` means
        the cache is valid; any dispatcher-level mutation (a foreign
        optimizer's add_, copy_, load) bumps it and triggers a refresh
        here instead of silently serving stale compute weights."""
        if self.weight_masked is None:
            return None
        if self.weight._version != self._w_version:
            self.refresh_cache()
        return self.weight_masked

    def refresh_cache(self) -> None:
        if self.compute_dtype is None:
            return
        self._w_version = self.weight._version
        with torch.no_grad():
            wm = TF.mask_apply(self.weight, self.mask, self.compute_dtype)
        if self.weight_masked is not None and \
                self.weight_masked.shape == wm.shape and \
                self.weight_masked.dtype == wm.dtype and \
                self.weight_masked.device == wm.device:
            self.weight_masked.copy_(wm)
        else:
            self.weight_masked = wm

    def _load_from_state_dict(self, *args, **kwargs):
        super()._load_from_state_dict(*args, **kwargs)
        self.refresh_cache()

    # sparsity accounting
    @torch.no_grad()
    def sparsity(self) -> float:
        return float((self.mask == 0).sum().item()) / self.mask.numel()


def _bias_like(bias: Optional[torch.Tensor], w: torch.Tensor):
    if bias is not None and bias.dtype != w.dtype \
            and not torch.is_autocast_enabled():
        return bias.to(w.dtype)
    return bias


class ConvMask(_MaskedMixin, nn.Conv2d):
    """Conv2d with a multiplicative 0/1 weight mask (reference:
    utils/mask_layers.py:10-43).

    1x1 stride-1 convolutions on channels_last GPU tensors are plain
    GEMMs (the NHWC reshape is free) and route through the in-house MFMA
    masked GEMM when ``TURBOPRUNE_CONV1X1=gemm``; other shapes go through
    the library conv on the cached masked weight."""

    def __init__(self, **kwargs) -> None:
        super().__init__(**kwargs)
        self._init_mask()

    def _gemm_1x1_ok(self, x: torch.Tensor) -> bool:
        import os
        return (os.environ.get("TURBOPRUNE_CONV1X1", "") == "gemm"
                and x.is_cuda and self.kernel_size == (1, 1)
                and self.stride == (1, 1) and self.padding == (0, 0)
                and self.groups == 1
                and x.is_contiguous(memory_format=torch.channels_last))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if self._gemm_1x1_ok(x):
            n, c, h, w_ = x.shape
            x2 = x.permute(0, 2, 3, 1).reshape(-1, c)  # free view on NHWC
            cache = self._fresh_cache()
            y2 = TF.masked_linear(
                x2, self.weight.reshape(self.out_channels, c),
                self.mask.reshape(self.out_channels, c), self.bias,
                cache.reshape(self.out_channels, c)
                if cache is not None else None,
                self.compute_dtype)
            return y2.view(n, h, w_, self.out_channels).permute(0, 3, 1, 2)
        from turboprune_amd.ops import conv_native
        cache = self._fresh_cache()
        if (cache is not None and cache.dtype == torch.bfloat16
                and x.is_cuda):
            # mirror autocast input-cast semantics for the custom Function
            if x.dtype != torch.bfloat16 and torch.is_autocast_enabled():
                x = x.to(torch.bfloat16)
            if conv_native.native_conv_ok(x, cache, self.stride,
                                          self.padding, self.dilation,
                                          self.groups) and (
                    os.environ.get("TURBOPRUNE_CONV", "auto") == "native"
                    or conv_native.AutoBackend.any_native(
                        self.out_channels, self.in_channels,
                        self.kernel_size[0], self.stride[0])):
                # one-node masked conv: fp32 master -> bf16 compute ->
                # fp32 mask⊙wrw grad (no bf16 grad rounding). Shapes
                # whose measured plan is all-library skip this Function
                # and take the plain conv path below (zero overhead).
                return conv_native.masked_conv2d_native(
                    x, self.weight, self.mask,
                    _bias_like(self.bias, cache), self.stride[0],
                    self.padding[0], cache)
        w = TF.masked_weight(self.weight, self.mask, cache,
                             self.compute_dtype)
        return torch.nn.functional.conv2d(
            x, w, _bias_like(self.bias, w), self.stride, self.padding,
            self.dilation, self.groups)


class LinearMask(_MaskedMixin, nn.Linear):
    """Linear with a multiplicative 0/1 weight mask (reference:
    utils/mask_layers.py:46-79)."""

    def __init__(self, **kwargs) -> None:
        super().__init__(**kwargs)
        self._init_mask()

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return TF.masked_linear(x, self.weight, self.mask, self.bias,
                                self._fresh_cache(), self.compute_dtype)


class Conv1dMask(_MaskedMixin, nn.Conv1d):
    """Linear-as-1x1-conv with weight mask (reference:
    utils/mask_layers.py:82-128). Weight shape (out, in, 1) as in the
    reference; forward evaluates the equivalent masked linear so (B, C)
    and (B, N, C) inputs both work."""

    def __init__(self, in_features: int, out_features: int,
                 bias: bool = False) -> None:
        super().__init__(in_channels=in_features, out_channels=out_features,
                         kernel_size=1, stride=1, bias=bias)
        self._init_mask()

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        cache = self._fresh_cache()
        return TF.masked_linear(
            x, self.weight.squeeze(-1), self.mask.squeeze(-1), self.bias,
            cache.squeeze(-1) if cache is not None else None,
            self.compute_dtype)


MASKED_LAYER_TYPES = (ConvMask, LinearMask, Conv1dMask)


def masked_modules(model: nn.Module):
    """Iterate (name, module) over all masked layers of a model."""
    for n, m in model.named_modules():
        if isinstance(m, MASKED_LAYER_TYPES):
            yield n, m
