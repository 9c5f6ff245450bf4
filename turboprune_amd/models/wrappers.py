"""PruneModel wrapper + model factory.

Mirrors the reference's wrapper surface (utils/custom_models.py):
``PruneModel`` holds the inner architecture as ``self.model`` and provides
sparsity accounting, weight rewinding (imp/wr/lrr semantics), mask
reset/load. ``TorchVisionModel``/``CustomModel`` equivalents collapse into
``build_model(cfg)``: torchvision is not in this stack, the architectures
live natively in turboprune_amd.models (same module naming), and the
Custom/DeiT path WORKS (the reference's is latent-broken).

Checkpoint format (must match the reference): ``torch.save`` of the inner
model's ``state_dict()`` — keys like ``conv1.weight``,
``layer1.0.conv1.mask``; masks are full fp32 tensors.
"""

from __future__ import annotations

import os
from typing import Any, Optional

import torch
import torch.nn as nn

from turboprune_amd.models import deit as deit_models
from turboprune_amd.models import resnet as resnet_models
from turboprune_amd.models import vgg as vgg_models
from turboprune_amd.ops.mask_layers import masked_modules

_FACTORIES = {}
for _mod in (resnet_models, vgg_models, deit_models):
    for _name in dir(_mod):
        _fn = getattr(_mod, _name)
        if callable(_fn) and (_name.startswith(
                ("resnet", "wide_resnet", "vgg", "local_deit"))):
            _FACTORIES[_name] = _fn


def available_models():
    return sorted(_FACTORIES)


class PruneModel(nn.Module):
    """Wrapper owning the inner architecture (``self.model``) plus
    mask/rewind utilities (reference: utils/custom_models.py:18-166)."""

    def __init__(self, model: nn.Module):
        super().__init__()
        self.model = model

    def forward(self, x):
        return self.model(x)

    # --- sparsity accounting ---------------------------------------------
    @torch.no_grad()
    def get_overall_sparsity(self) -> float:
        """Percent of zero mask entries over all masked layers
        (reference: custom_models.py:51-62)."""
        zeros = 0
        total = 0
        for _, m in masked_modules(self.model):
            zeros += int((m.mask == 0).sum().item())
            total += m.mask.numel()
        return 100.0 * zeros / max(total, 1)

    @torch.no_grad()
    def layer_sparsity(self):
        return [(n, 100.0 * m.sparsity()) for n, m in masked_modules(self.model)]

    def print_layer_sparsity(self) -> None:
        try:
            from rich.console import Console
            from rich.table import Table
            table = Table(title="Layer sparsity (%)")
            table.add_column("layer")
            table.add_column("sparsity", justify="right")
            for n, s in self.layer_sparsity():
                table.add_row(n, f"{s:.2f}")
            Console().print(table)
        except ImportError:
            for n, s in self.layer_sparsity():
                print(f"{n}: {s:.2f}%")

    # --- checkpoint plumbing ----------------------------------------------
    def load_model(self, load_path: str) -> None:
        state = torch.load(load_path, map_location="cpu", weights_only=True)
        self.model.load_state_dict(state)
        self.refresh_caches()

    @torch.no_grad()
    def reset_weights(self, cfg: Any, expt_dir: str) -> None:
        """Rewind weights per training type: imp -> model_init.pt, wr ->
        model_rewind.pt, lrr/at_init -> no-op. Only non-mask keys with
        matching shapes are copied (reference: custom_models.py:112-146)."""
        training_type = cfg.pruning_params.training_type
        if training_type == "imp":
            checkpoint_file = "model_init.pt"
        elif training_type == "wr":
            checkpoint_file = "model_rewind.pt"
        else:
            return
        original = torch.load(os.path.join(expt_dir, "checkpoints",
                                           checkpoint_file),
                              map_location="cpu", weights_only=True)
        current = self.model.state_dict()
        for name, param in original.items():
            if (name in current and current[name].shape == param.shape
                    and not name.endswith("mask")):
                current[name].copy_(param.to(current[name].device))
        self.model.load_state_dict(current)
        self.refresh_caches()

    @torch.no_grad()
    def reset_masks(self) -> None:
        for _, m in masked_modules(self.model):
            m.mask.fill_(1)
            m.refresh_cache()

    @torch.no_grad()
    def load_only_masks(self, load_path: str) -> None:
        original = torch.load(load_path, map_location="cpu", weights_only=True)
        current = self.model.state_dict()
        for name, param in original.items():
            if (name in current and current[name].shape == param.shape
                    and name.endswith("mask")):
                current[name].copy_(param.to(current[name].device))
        self.model.load_state_dict(current)
        self.refresh_caches()

    # --- masked-weight caches ---------------------------------------------
    def enable_caches(self, compute_dtype: torch.dtype) -> None:
        for _, m in masked_modules(self.model):
            m.enable_cache(compute_dtype)

    def disable_caches(self) -> None:
        for _, m in masked_modules(self.model):
            m.disable_cache()

    def refresh_caches(self) -> None:
        for _, m in masked_modules(self.model):
            m.refresh_cache()


def num_classes_of(dataset_name: str) -> int:
    return {"CIFAR10": 10, "CIFAR100": 100, "ImageNet": 1000}[dataset_name]


def build_model(cfg: Any, num_classes: Optional[int] = None) -> PruneModel:
    """Construct the masked model named by ``cfg.model_params.model_name``
    (resnet*/vgg*/local_deit_*), with CIFAR stem surgery for CIFAR
    datasets (reference: custom_models.py:169-245, fixed DeiT path)."""
    name = cfg.model_params.model_name
    dataset = cfg.dataset_params.dataset_name
    if num_classes is None:
        num_classes = num_classes_of(dataset)
    # short aliases for the timm-registered local_deit_* names
    # (reference surface: utils/deit.py:69-253)
    if name not in _FACTORIES:
        for prefix in ("local_", ""):
            cand = f"{prefix}{name}_patch16_224"
            if cand in _FACTORIES:
                name = cand
                break
        else:
            if f"local_{name}" in _FACTORIES:
                name = f"local_{name}"
    if name not in _FACTORIES:
        raise ValueError(f"unknown model '{name}'; available: "
                         f"{available_models()}")
    factory = _FACTORIES[name]
    kwargs = {"num_classes": num_classes}
    if name.startswith(("resnet", "vgg")):
        kwargs["cifar_stem"] = dataset in ("CIFAR10", "CIFAR100")
    inner = factory(**kwargs)
    return PruneModel(inner)
