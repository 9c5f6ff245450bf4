"""Pruning scorers vs reference closed forms."""

import math

import pytest
import torch
import torch.nn as nn

from turboprune_amd.config import compose
from turboprune_amd.ops.mask_layers import ConvMask, LinearMask, masked_modules
from turboprune_amd.pruning import (balanced_keep_probabilities,
                                    erk_keep_probabilities, prune_er_balanced,
                                    prune_er_erk, prune_mag,
                                    prune_random_balanced, prune_random_erk,
                                    prune_snip, prune_synflow,
                                    prune_the_model)
from turboprune_amd.utils.experiment import (generate_cyclical_schedule,
                                             generate_densities)


def tiny_model():
    torch.manual_seed(0)
    return nn.Sequential(
        ConvMask(in_channels=3, out_channels=8, kernel_size=3, padding=1,
                 bias=False),
        nn.ReLU(),
        nn.AdaptiveAvgPool2d(1),
        nn.Flatten(),
        LinearMask(in_features=8, out_features=4, bias=True),
    )


def overall_density(model):
    kept = sum(int(m.mask.sum()) for _, m in masked_modules(model))
    total = sum(m.mask.numel() for _, m in masked_modules(model))
    return kept / total


def test_prune_mag_matches_kthvalue_oracle():
    model = tiny_model()
    density = 0.5
    scores = torch.cat([(m.mask * m.weight).abs().flatten()
                        for _, m in masked_modules(model)])
    k = int((1 - density) * scores.numel())
    thr = torch.kthvalue(scores, k).values.item()
    prune_mag(model, density)
    for _, m in masked_modules(model):
        expected = ((m.mask_score if hasattr(m, 'mask_score') else
                     (m.weight.abs() > thr))).float()
        # every kept weight is above threshold, every dropped <= threshold
        kept = m.mask == 1
        assert torch.all(m.weight.abs()[kept] > thr)
        assert torch.all(m.weight.abs()[~kept] <= thr)


def test_prune_mag_density_reached():
    model = tiny_model()
    prune_mag(model, 0.3)
    assert overall_density(model) == pytest.approx(0.3, abs=0.02)


def test_iterative_pruning_is_monotone():
    model = tiny_model()
    prune_mag(model, 0.5)
    masks_before = {n: m.mask.clone() for n, m in masked_modules(model)}
    prune_mag(model, 0.25)
    for n, m in masked_modules(model):
        # nothing pruned before is ever revived
        assert torch.all(m.mask[masks_before[n] == 0] == 0)


def test_erk_probabilities_closed_form():
    model = tiny_model()
    density = 0.4
    probs = erk_keep_probabilities(model, density)
    ratios, numels = [], []
    for _, m in masked_modules(model):
        ratios.append(sum(m.weight.shape) / m.weight.numel())
        numels.append(m.weight.numel())
    C = density * sum(numels) / sum(r * n for r, n in zip(ratios, numels))
    for p, r in zip(probs, ratios):
        assert p == pytest.approx(min(max(C * r, 0.0), 1.0))
    # expected kept ≈ density (when nothing clamps)
    exp_kept = sum(p * n for p, n in zip(probs, numels)) / sum(numels)
    if all(p < 1.0 for p in probs):
        assert exp_kept == pytest.approx(density, rel=1e-6)


def overflow_model():
    """Small layer FIRST so balanced overflow redistributes forward
    (the reference's correction only reaches later layers,
    pruning_utils.py:320-326)."""
    torch.manual_seed(0)
    return nn.Sequential(
        LinearMask(in_features=4, out_features=4, bias=False),     # 16 params
        nn.ReLU(),
        LinearMask(in_features=4, out_features=256, bias=False),   # 1024
    )


def test_balanced_probabilities_overflow_redistribution():
    model = overflow_model()
    density = 0.9  # X = 468 > 16 for the first layer -> overflow
    probs = balanced_keep_probabilities(model, density)
    # reference redistribution (pruning_utils.py:320-326): the overflow
    # diff is divided by (L - l) with l the CURRENT 0-based index, so only
    # diff/2 reaches the second layer here
    X = 0.9 * (16 + 1024) / 2            # 468
    X2 = X + (X - 16) / 2                # 694
    assert probs[0] == 1.0
    assert probs[1] == pytest.approx(X2 / 1024)
    assert max(probs) <= 1.0


def bigger_model():
    """Layers large enough that ERK probabilities don't clamp at 1, so
    expected kept == requested density."""
    torch.manual_seed(0)
    return nn.Sequential(
        ConvMask(in_channels=32, out_channels=32, kernel_size=3,
                 padding=1, bias=False),
        nn.ReLU(),
        nn.AdaptiveAvgPool2d(1),
        nn.Flatten(),
        LinearMask(in_features=32, out_features=64, bias=False),
    )


def _expected_density(model, probs):
    numels = [m.weight.numel() for _, m in masked_modules(model)]
    return sum(p * n for p, n in zip(probs, numels)) / sum(numels)


def test_er_variants_hit_their_budget():
    """Realized density matches the per-layer budget (the budget itself
    may differ from the request when ERK clamps — reference behavior)."""
    torch.manual_seed(7)
    for fn, budget in ((prune_er_erk, erk_keep_probabilities),
                       (prune_er_balanced, balanced_keep_probabilities)):
        model = bigger_model()
        expected = _expected_density(model, budget(model, 0.5))
        fn(model, 0.5)
        assert overall_density(model) == pytest.approx(expected, abs=0.03)


def test_random_variants_hit_their_budget():
    torch.manual_seed(8)
    for fn, budget in ((prune_random_erk, erk_keep_probabilities),
                       (prune_random_balanced, balanced_keep_probabilities)):
        model = bigger_model()
        expected = _expected_density(model, budget(model, 0.5))
        fn(model, 0.5)
        assert overall_density(model) == pytest.approx(expected, abs=0.03)


def _fake_loader():
    x = torch.randn(8, 3, 8, 8)
    y = torch.randint(0, 4, (8,))
    return [(x, y)]


def test_snip_runs_and_prunes():
    model = tiny_model()
    prune_snip(model, 0.5, _fake_loader(), torch.device("cpu"),
               amp_dtype=torch.float32)
    assert overall_density(model) == pytest.approx(0.5, abs=0.05)
    # grads cleared afterwards
    assert all(p.grad is None for p in model.parameters())


def test_synflow_runs_prunes_and_restores_signs():
    model = tiny_model()
    w_before = {n: m.weight.detach().clone()
                for n, m in masked_modules(model)}
    prune_synflow(model, 0.5, _fake_loader(), torch.device("cpu"))
    assert overall_density(model) == pytest.approx(0.5, abs=0.05)
    for n, m in masked_modules(model):
        assert torch.allclose(m.weight.detach(), w_before[n])


def test_prune_the_model_dispatch():
    cfg = compose("cifar10_er_erk")
    model = tiny_model()
    prune_the_model(cfg, model, 0.5)
    assert overall_density(model) < 1.0


def test_density_ladder_geometric():
    cfg = compose("imagenet_imp")  # prune_rate .2, target .999
    densities = generate_densities(cfg, 0.0)
    assert densities[0] == 1.0
    for a, b in zip(densities, densities[1:]):
        assert b == pytest.approx(a * 0.8)
    assert densities[-1] <= 1 - 0.999 + 1e-9 or \
        densities[-1] == pytest.approx((0.8) ** (len(densities) - 1))
    # last density is the first at/below target
    assert densities[-2] > 0.001 >= densities[-1] * (1 + 1e-9) or \
        densities[-1] <= 0.001


def test_density_ladder_pai_single_level():
    cfg = compose("cifar10_er_erk", ["pruning_params.target_sparsity=0.9"])
    assert generate_densities(cfg, 0.0) == [pytest.approx(0.1)]


def test_cyclic_schedule_budget():
    for strat in ("constant", "linear_increase", "linear_decrease",
                  "exponential_increase", "exponential_decrease",
                  "cyclic_peak", "alternating", "plateau"):
        cfg = compose("cifar10_er_erk", [
            "experiment_params.epochs_per_level=60",
            "+cyclic_training.num_cycles=5",
            f"+cyclic_training.strategy={strat}"])
        cfg.cyclic_training.num_cycles = 5
        cfg.cyclic_training.strategy = strat
        epochs = generate_cyclical_schedule(cfg)
        assert len(epochs) == 5
        assert sum(epochs) <= 60
        assert all(e >= 0 for e in epochs)


def test_cyclic_schedule_single_cycle():
    cfg = compose("cifar10_er_erk", ["experiment_params.epochs_per_level=7"])
    assert generate_cyclical_schedule(cfg) == [7]
