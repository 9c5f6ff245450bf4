"""Property-based tests (hypothesis) for the math-heavy components:
density ladders, cyclic schedules, budget formulas, config composition."""

import hypothesis.strategies as st
import pytest
import torch
from hypothesis import given, settings

from turboprune_amd.config import compose
from turboprune_amd.utils.experiment import (generate_cyclical_schedule,
                                             generate_densities)

STRATEGIES = ["linear_increase", "linear_decrease", "exponential_decrease",
              "exponential_increase", "cyclic_peak", "alternating",
              "plateau", "constant"]


@settings(max_examples=60, deadline=None)
@given(prune_rate=st.floats(0.05, 0.8),
       target=st.floats(0.1, 0.999),
       current=st.floats(0.0, 0.5))
def test_density_ladder_properties(prune_rate, target, current):
    cfg = compose("imagenet_imp", [
        f"pruning_params.prune_rate={prune_rate}",
        f"pruning_params.target_sparsity={target}"])
    densities = generate_densities(cfg, current)
    assert len(densities) >= 1
    # strictly decreasing geometric ladder
    for a, b in zip(densities, densities[1:]):
        assert b == pytest.approx(a * (1 - prune_rate), rel=1e-9)
    # ends at/below the target density; the PREVIOUS level was above it
    assert densities[-1] <= (1 - target) + 1e-12
    if len(densities) > 1:
        assert densities[-2] > (1 - target)
    assert densities[0] == pytest.approx(1 - current)


@settings(max_examples=80, deadline=None)
@given(epochs=st.integers(1, 300), cycles=st.integers(1, 12),
       strategy=st.sampled_from(STRATEGIES))
def test_cyclic_schedule_properties(epochs, cycles, strategy):
    cfg = compose("cifar10_er_erk", [
        f"experiment_params.epochs_per_level={epochs}"])
    cfg.cyclic_training.num_cycles = cycles
    cfg.cyclic_training.strategy = strategy
    sched = generate_cyclical_schedule(cfg)
    if cycles == 1:
        assert sched == [epochs]
        return
    assert len(sched) == cycles
    assert sum(sched) <= epochs  # never exceeds the budget
    assert all(isinstance(e, int) for e in sched)


@settings(max_examples=40, deadline=None)
@given(lr=st.floats(1e-4, 1.0), wd=st.floats(0.0, 0.1),
       seed=st.integers(0, 10_000))
def test_override_roundtrip(lr, wd, seed):
    cfg = compose("cifar10_er_erk", [
        f"optimizer_params.lr={lr}",
        f"optimizer_params.weight_decay={wd}",
        f"experiment_params.seed={seed}"])
    assert cfg.optimizer_params.lr == pytest.approx(lr)
    assert cfg.optimizer_params.weight_decay == pytest.approx(wd)
    assert cfg.experiment_params.seed == seed


@settings(max_examples=30, deadline=None)
@given(n=st.integers(2, 2000), seed=st.integers(0, 1000))
def test_kth_cpu_matches_sort(n, seed):
    from turboprune_amd.ops.functional import kth_smallest
    g = torch.Generator().manual_seed(seed)
    v = torch.randn(n, generator=g)
    k = 1 + seed % n
    assert kth_smallest(v, k) == torch.sort(v).values[k - 1].item()


@settings(max_examples=30, deadline=None)
@given(density=st.floats(0.05, 0.95), seed=st.integers(0, 100))
def test_erk_budget_invariants(density, seed):
    import torch.nn as nn

    from turboprune_amd.ops.mask_layers import LinearMask
    from turboprune_amd.pruning import (balanced_keep_probabilities,
                                        erk_keep_probabilities)
    torch.manual_seed(seed)
    model = nn.Sequential(
        LinearMask(in_features=16, out_features=32, bias=False),
        LinearMask(in_features=32, out_features=8, bias=False),
        LinearMask(in_features=8, out_features=64, bias=False),
    )
    for fn in (erk_keep_probabilities, balanced_keep_probabilities):
        probs = fn(model, density)
        assert len(probs) == 3
        assert all(0.0 <= p <= 1.0 for p in probs)
