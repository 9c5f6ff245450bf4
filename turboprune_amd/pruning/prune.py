"""Pruning engine — the scientific core.

Reference-exact semantics (utils/pruning_utils.py): every
scorer builds per-layer score tensors over masked layers, thresholds via
a GLOBAL k-th-smallest (or per-layer for the random_* variants), and
rewrites masks as ``where(score <= threshold, 0, 1)``. Scores always
include the current mask as a factor, so iterative pruning is monotone.
ER variants compute per-layer keep-probabilities and Bernoulli-fill.

MI355X-native path: scores are built by fused elementwise HIP kernels,
the threshold comes from a radix-select k-th-value kernel (no full sort,
single GPU, training precision — reference README.md:69), Bernoulli fills
use a Philox HIP kernel. Pruning runs on rank 0 only; propagation to
other ranks is an explicit RCCL broadcast at harness (re)construction
(reference mechanism: DDP re-wrap, run_experiment.py:113-115).

Device handling honors the model's device (the reference hardcodes
``cuda`` in SNIP/SynFlow — defect §2.6.7 — fixed here).
"""

from __future__ import annotations

from typing import Any, Iterable, List, Optional

import torch
import torch.nn as nn

from turboprune_amd.ops import functional as TF
from turboprune_amd.ops.mask_layers import masked_modules


def _model_device(model: nn.Module) -> torch.device:
    return next(model.parameters()).device


@torch.no_grad()
def _threshold_global(scores: List[torch.Tensor], density: float) -> Optional[float]:
    """Global k-th smallest over concatenated scores;
    k = int((1-density)*numel) (reference: pruning_utils.py:77-79)."""
    flat = torch.cat([s.reshape(-1) for s in scores])
    k = int((1.0 - density) * flat.numel())
    if k < 1:
        return None
    return TF.kth_smallest(flat, k)


@torch.no_grad()
def _rewrite_masks(model: nn.Module, scores: dict, threshold: float) -> None:
    for n, m in masked_modules(model):
        TF.mask_from_threshold_(m.mask, scores[n].to(m.mask.device), threshold)
        m.refresh_cache()


@torch.no_grad()
def prune_mag(model: nn.Module, density: float) -> nn.Module:
    """Magnitude pruning: score = |mask * weight|
    (reference: pruning_utils.py:61-89)."""
    scores = {n: TF.masked_abs_score(m.weight, m.mask)
              for n, m in masked_modules(model)}
    thr = _threshold_global(list(scores.values()), density)
    if thr is not None:
        _rewrite_masks(model, scores, thr)
    return model


def prune_snip(model: nn.Module, density: float,
               dataloader: Iterable, device: Optional[torch.device] = None,
               amp_dtype: torch.dtype = torch.bfloat16) -> nn.Module:
    """SNIP: score = |grad * weight * mask| after ONE cross-entropy
    backward on ONE training batch under autocast
    (reference: pruning_utils.py:160-205)."""
    device = device or _model_device(model)
    model.zero_grad(set_to_none=True)
    was_training = model.training
    model.train()
    inputs, targets = next(iter(dataloader))
    inputs = inputs.to(device, non_blocking=True)
    targets = targets.to(device, non_blocking=True)
    use_amp = device.type == "cuda" and amp_dtype != torch.float32
    with torch.autocast(device_type="cuda", dtype=amp_dtype, enabled=use_amp):
        out = model(inputs)
        loss = TF.cross_entropy(out, targets)
    loss.backward()

    with torch.no_grad():
        scores = {}
        for n, m in masked_modules(model):
            g = m.weight.grad
            if g is None:
                g = torch.zeros_like(m.weight)
            scores[n] = TF.masked_abs_score(m.weight, m.mask, g.float())
        thr = _threshold_global(list(scores.values()), density)
        if thr is not None:
            _rewrite_masks(model, scores, thr)
    model.zero_grad(set_to_none=True)
    model.train(was_training)
    return model


def prune_synflow(model: nn.Module, density: float,
                  dataloader: Iterable,
                  device: Optional[torch.device] = None,
                  amp_dtype: torch.dtype = torch.bfloat16) -> nn.Module:
    """SynFlow: linearize params to |param| (signs saved), backward of
    sum(model(ones_input)), score = |mask * grad * weight|, restore signs
    (reference: pruning_utils.py:208-285). The forward runs in the
    model's CURRENT train/eval mode and under autocast, exactly as the
    reference does (BN batch stats / running-stat updates included)."""
    device = device or _model_device(model)

    # linearize: theta <- |theta| over ALL params & buffers, keep signs
    # (fused HIP pass on GPU — K10)
    signs = {}
    with torch.no_grad():
        state = model.state_dict()
        for name, t in state.items():
            if t.is_floating_point():
                signs[name] = TF.synflow_linearize_(t)

    inputs, _ = next(iter(dataloader))
    input_shape = list(inputs.shape)
    input_shape[0] = 1
    ones = torch.ones(input_shape, device=device)

    model.zero_grad(set_to_none=True)
    use_amp = device.type == "cuda" and amp_dtype != torch.float32
    with torch.autocast(device_type="cuda", dtype=amp_dtype,
                        enabled=use_amp):
        out = model(ones)
        if isinstance(out, tuple):
            out = out[0]
        torch.sum(out).backward()

    with torch.no_grad():
        scores = {}
        for n, m in masked_modules(model):
            g = m.weight.grad
            if g is None:
                g = torch.zeros_like(m.weight)
            scores[n] = TF.masked_abs_score(m.weight, m.mask, g.float())
        # restore signs
        state = model.state_dict()
        for name, t in state.items():
            if name in signs:
                TF.synflow_restore_(t, signs[name])
        thr = _threshold_global(list(scores.values()), density)
        if thr is not None:
            _rewrite_masks(model, scores, thr)
    model.zero_grad(set_to_none=True)
    return model


@torch.no_grad()
def erk_keep_probabilities(model: nn.Module, density: float) -> List[float]:
    """ERK layer budgets: keep-prob ∝ sum(weight.shape)/numel, scaled so the
    total kept = density, clamped to [0,1]
    (reference: pruning_utils.py:109-127,350-378)."""
    ratios, numels = [], []
    for _, m in masked_modules(model):
        ratios.append(float(sum(m.weight.shape)) / m.weight.numel())
        numels.append(m.weight.numel())
    total = sum(numels)
    kept = sum(r * n for r, n in zip(ratios, numels))
    C = density * total / kept
    return [min(max(C * r, 0.0), 1.0) for r in ratios]


@torch.no_grad()
def balanced_keep_probabilities(model: nn.Module, density: float) -> List[float]:
    """Balanced layer budgets: equal param count X = density*total/L per
    layer with overflow redistribution
    (reference: pruning_utils.py:298-327,381-415)."""
    numels = [m.weight.numel() for _, m in masked_modules(model)]
    L = len(numels)
    total = sum(numels)
    X = density * total / L
    probs = []
    for i, n in enumerate(numels):
        if X / n < 1.0:
            probs.append(X / n)
        else:
            probs.append(1.0)
            diff = X - n
            X = X + diff / (L - i)
    return probs


@torch.no_grad()
def _prune_random(model: nn.Module, probs: List[float]) -> nn.Module:
    """Per-layer random scores |mask * randn|, per-layer k-th-value
    threshold at each layer's keep-prob (reference:
    pruning_utils.py:92-146,288-347)."""
    for (n, m), keep in zip(masked_modules(model), probs):
        score = TF.masked_abs_score(m.weight, m.mask,
                                    torch.randn_like(m.weight))
        flat = score.reshape(-1)
        k = int((1.0 - keep) * flat.numel())
        thr = 0.0 if k == 0 else TF.kth_smallest(flat, k)
        TF.mask_from_threshold_(m.mask, score, thr)
        m.refresh_cache()
    return model


@torch.no_grad()
def prune_random_erk(model: nn.Module, density: float) -> nn.Module:
    return _prune_random(model, erk_keep_probabilities(model, density))


@torch.no_grad()
def prune_random_balanced(model: nn.Module, density: float) -> nn.Module:
    return _prune_random(model, balanced_keep_probabilities(model, density))


@torch.no_grad()
def prune_er_erk(model: nn.Module, density: float) -> nn.Module:
    for (n, m), p in zip(masked_modules(model),
                         erk_keep_probabilities(model, density)):
        m.set_er_mask(p)
    return model


@torch.no_grad()
def prune_er_balanced(model: nn.Module, density: float) -> nn.Module:
    for (n, m), p in zip(masked_modules(model),
                         balanced_keep_probabilities(model, density)):
        m.set_er_mask(p)
    return model


def prune_the_model(cfg: Any, model: nn.Module, target_density: float,
                    dataloader: Optional[Iterable] = None,
                    device: Optional[torch.device] = None) -> nn.Module:
    """Dispatch to the configured scorer (reference:
    pruning_utils.py:23-58). `model` is the INNER masked model (callers
    unwrap DDP/PruneModel first). Data-driven methods (snip/synflow) need
    `dataloader`."""
    method = cfg.pruning_params.prune_method
    if method == "just dont":
        return model
    if method == "mag":
        return prune_mag(model, target_density)
    if method == "snip":
        assert dataloader is not None, "snip needs a train dataloader"
        amp = {"bfloat16": torch.bfloat16, "float16": torch.float16,
               "float32": torch.float32}[
                   cfg.select("experiment_params.training_precision",
                              "bfloat16")]
        return prune_snip(model, target_density, dataloader, device, amp)
    if method == "synflow":
        assert dataloader is not None, "synflow needs a train dataloader"
        amp = {"bfloat16": torch.bfloat16, "float16": torch.float16,
               "float32": torch.float32}[
                   cfg.select("experiment_params.training_precision",
                              "bfloat16")]
        return prune_synflow(model, target_density, dataloader, device, amp)
    if method == "random_erk":
        return prune_random_erk(model, target_density)
    if method == "random_balanced":
        return prune_random_balanced(model, target_density)
    if method == "er_erk":
        return prune_er_erk(model, target_density)
    if method == "er_balanced":
        return prune_er_balanced(model, target_density)
    raise ValueError(f"Unknown pruning method: {method}")
