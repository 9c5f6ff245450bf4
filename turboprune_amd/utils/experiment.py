"""Experiment management: directory layout, seeds, density ladders, schedules.

Formats match the reference exactly so artifacts are drop-in compatible:

- experiment dir ``{base_dir}/{prefix}__{uuid6}__{YYYYmmdd_HHMMSS}/`` with
  subdirs ``checkpoints/``, ``metrics/``, ``metrics/level_wise_metrics/``,
  ``artifacts/`` (reference: utils/harness_utils.py:49-94);
- geometric density ladder ``(1-prune_rate)^i`` for iterative methods,
  single-level for prune-at-init (reference: utils/harness_utils.py:117-145);
- cyclic epoch schedules with total-budget rescaling (reference:
  utils/harness_utils.py:159-245);
- ``save_model`` stores the INNER model's state_dict (keys like
  ``conv1.weight``, ``layer1.0.conv1.mask`` — no wrapper prefix;
  reference: utils/harness_utils.py:354-365).

``resume_experiment`` works (the reference's is broken — SURVEY §2.6.2).
"""

from __future__ import annotations

import os
import random
import uuid
from datetime import datetime
from typing import Any, List, Tuple

import numpy as np
import torch


def gen_expt_dir(cfg: Any) -> Tuple[str, str]:
    """Create the experiment directory tree; returns (prefix, expt_dir)."""
    base_dir = cfg.experiment_params.base_dir
    num_cycles = int(cfg.select("cyclic_training.num_cycles", 1))

    prefix = (
        f"{cfg.dataset_params.dataset_name}"
        f"_model_{cfg.model_params.model_name}"
        f"_trainingtype_{cfg.pruning_params.training_type}"
        f"_prunemethod_{cfg.pruning_params.prune_method}"
        f"_target_{float(cfg.pruning_params.target_sparsity):.2f}"
        f"_seed_{cfg.experiment_params.seed}"
        f"_budget_{cfg.experiment_params.epochs_per_level}epochs"
        + (
            f"_cycles_{num_cycles}_strat_{cfg.cyclic_training.strategy}"
            if num_cycles > 1
            else ""
        )
        + f"_lr_{float(cfg.optimizer_params.lr):.3f}"
        + f"_mom_{float(cfg.optimizer_params.momentum):.1f}"
        + f"_wd_{float(cfg.optimizer_params.weight_decay):.4f}"
        + f"_sched_{cfg.optimizer_params.scheduler_type}"
    )

    current_time = datetime.now().strftime("%Y%m%d_%H%M%S")
    unique_id = uuid.uuid4().hex[:6]
    expt_dir = os.path.join(base_dir, f"{prefix}__{unique_id}__{current_time}")

    os.makedirs(expt_dir, exist_ok=True)
    for subdir in ("checkpoints", "metrics", "metrics/level_wise_metrics",
                   "artifacts"):
        os.makedirs(os.path.join(expt_dir, subdir), exist_ok=True)
    return prefix, expt_dir


def set_seed(cfg: Any, is_deterministic: bool = False) -> None:
    seed = int(cfg.experiment_params.seed)
    np.random.seed(seed)
    random.seed(seed)
    torch.manual_seed(seed)
    if torch.cuda.is_available():
        torch.cuda.manual_seed_all(seed)
    if is_deterministic:
        torch.backends.cudnn.deterministic = True
        torch.backends.cudnn.benchmark = False
        torch.use_deterministic_algorithms(True)
    os.environ["PYTHONHASHSEED"] = str(seed)


def generate_densities(cfg: Any, current_sparsity: float = 0.0) -> List[float]:
    """Density per pruning level (reference math, utils/harness_utils.py:117-145).

    Iterative methods walk the geometric ladder ``d_{i+1} = d_i*(1-prune_rate)``
    from (1 - current_sparsity) down past (1 - target_sparsity), inclusive of
    the first density at/below target. PaI methods prune once to target.
    """
    method = cfg.pruning_params.prune_method
    target_sparsity = float(cfg.pruning_params.target_sparsity)

    if method in ("mag", "random_erk", "random_balanced"):
        prune_rate = float(cfg.pruning_params.prune_rate)
        densities: List[float] = []
        density = 1.0 - float(current_sparsity)
        target_density = 1.0 - target_sparsity
        while density > target_density:
            densities.append(density)
            density *= 1.0 - prune_rate
        densities.append(density)
        return densities
    if method in ("er_erk", "er_balanced", "synflow", "snip"):
        return [1.0 - target_sparsity]
    if method == "just dont":
        return [1.0]
    raise ValueError(f"Unknown pruning method: {method}")


def generate_cyclical_schedule(cfg: Any) -> List[int]:
    """Epochs per cycle for cyclic training (reference math,
    utils/harness_utils.py:159-245), including the budget rescale."""
    epochs_per_level = int(cfg.experiment_params.epochs_per_level)
    num_cycles = int(cfg.select("cyclic_training.num_cycles", 1))
    strategy = cfg.select("cyclic_training.strategy", "constant")

    if num_cycles <= 1:
        return [epochs_per_level]

    n = num_cycles
    if strategy == "linear_decrease":
        step = epochs_per_level / (n * (n + 1) / 2)
        epochs = [int(step * (n - i)) for i in range(n)]
    elif strategy == "linear_increase":
        step = epochs_per_level / (n * (n + 1) / 2)
        epochs = [int(step * (i + 1)) for i in range(n)]
    elif strategy == "exponential_decrease":
        factor = 0.5 ** (1 / (n - 1))
        total = sum(factor ** i for i in range(n))
        epochs = [int(epochs_per_level * factor ** i / total) for i in range(n)]
    elif strategy == "exponential_increase":
        factor = 2 ** (1 / (n - 1))
        total = sum(factor ** i for i in range(n))
        epochs = [int(epochs_per_level * factor ** i / total) for i in range(n)]
    elif strategy == "cyclic_peak":
        mid = n // 2
        inc = epochs_per_level / (mid * (mid + 1) / 2)
        dec = epochs_per_level / ((n - mid) * (n - mid + 1) / 2)
        epochs = [int(inc * (i + 1)) for i in range(mid)]
        epochs += [int(dec * (n - i)) for i in range(mid, n)]
    elif strategy == "alternating":
        high = epochs_per_level // (n // 2 + n % 2)
        low = epochs_per_level // (2 * (n // 2 + n % 2))
        epochs = [high if i % 2 == 0 else low for i in range(n)]
    elif strategy == "plateau":
        inc_cycles = n // 2
        plateau_cycles = n - inc_cycles
        inc = epochs_per_level / (inc_cycles * (inc_cycles + 1) / 2)
        epochs = [int(inc * (i + 1)) for i in range(inc_cycles)]
        epochs += [epochs_per_level // n for _ in range(plateau_cycles)]
    elif strategy == "constant":
        epochs = [epochs_per_level // n for _ in range(n)]
    else:
        raise ValueError(f"Unknown cyclic strategy: {strategy}")

    total = sum(epochs)
    if total > epochs_per_level:
        scale = epochs_per_level / total
        epochs = [int(e * scale) for e in epochs]
        excess = sum(epochs) - epochs_per_level
        if excess > 0:
            per = excess // len(epochs)
            rem = excess % len(epochs)
            epochs = [e - per for e in epochs]
            for i in range(rem):
                epochs[i] -= 1
    return epochs


def unwrap_inner_model(model: torch.nn.Module) -> torch.nn.Module:
    """Peel DDP/.compile/PruneModel wrappers down to the inner architecture
    whose state_dict defines the checkpoint format."""
    m = model
    for attr in ("_orig_mod", "module", "model"):
        while hasattr(m, attr) and isinstance(getattr(m, attr), torch.nn.Module):
            m = getattr(m, attr)
    return m


def save_model(model: torch.nn.Module, save_path: str,
               distributed: bool = False) -> None:
    """Save the inner model's state_dict — fp32 ``*.mask`` buffers included
    (checkpoint format, reference: utils/harness_utils.py:354-365)."""
    inner = unwrap_inner_model(model)
    torch.save(inner.state_dict(), save_path)


def resume_experiment(cfg: Any) -> Tuple[str, str, int]:
    """Resume at level granularity. Returns (prefix, expt_dir, resume_level).

    Expects ``experiment_params.resume_experiment_stuff.{resume_level,
    resume_expt_name}``; the previous level's checkpoint must exist for
    iterative training types. (The reference's version is broken:
    wrong arity and unbound prefix — SURVEY §2.6.2.)
    """
    stuff = cfg.select("experiment_params.resume_experiment_stuff", None)
    if stuff is None:
        raise ValueError("resume_experiment requires "
                         "experiment_params.resume_experiment_stuff")
    resume_level = int(stuff["resume_level"])
    resume_expt_name = stuff["resume_expt_name"]
    base_dir = cfg.experiment_params.base_dir

    expt_dir = os.path.join(base_dir, resume_expt_name)
    if not os.path.isdir(expt_dir):
        raise FileNotFoundError(f"experiment dir not found: {expt_dir}")
    # prefix is the dir name minus __uuid__timestamp
    prefix = os.path.basename(expt_dir).split("__")[0]

    if resume_level > 0 and cfg.pruning_params.training_type in ("imp", "wr", "lrr"):
        ckpt = os.path.join(expt_dir, "checkpoints",
                            f"model_level_{resume_level - 1}.pt")
        if not os.path.exists(ckpt):
            raise FileNotFoundError(
                f"previous level checkpoint not found at {ckpt}")
    return prefix, expt_dir, resume_level
