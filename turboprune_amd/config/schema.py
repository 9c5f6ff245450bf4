"""Config schema — actually enforced, unlike the reference.

The reference declares a dataclass schema (utils/harness_params.py:6-101)
but never registers it with hydra's ConfigStore, so nothing is validated
(reference defect). Here `validate_config` runs on every
composed config before an experiment starts.

Group/key names mirror the reference exactly.
"""

from __future__ import annotations

from typing import Any, List

PRUNE_METHODS = (
    "er_erk", "er_balanced", "random_erk", "random_balanced",
    "synflow", "snip", "mag", "just dont",
)
TRAINING_TYPES = ("imp", "wr", "lrr", "at_init")
SCHEDULER_TYPES = (
    "MultiStepLRWarmup", "ImageNetLRDropsWarmup", "TriangularSchedule",
    "ScheduleFree", "TrapezoidalSchedule", "OneCycleLR",
)
DATASETS = ("CIFAR10", "CIFAR100", "ImageNet")
DATALOADER_TYPES = ("torch", "airbench", "ffcv", "webdataset", "synthetic", "native")
MASK_LAYER_TYPES = ("ConvMask", "LinearMask")
PRECISIONS = ("bfloat16", "float16", "float32")
CYCLIC_STRATEGIES = (
    "linear_increase", "linear_decrease", "exponential_decrease",
    "exponential_increase", "cyclic_peak", "alternating", "plateau", "constant",
)

# methods that use the geometric density ladder and need prune_rate
ITERATIVE_METHODS = ("mag", "random_erk", "random_balanced")
PAI_METHODS = ("er_erk", "er_balanced", "synflow", "snip")


class ConfigError(ValueError):
    pass


def _req(cfg: Any, group: str, key: str, errors: List[str], typ=None):
    val = cfg.select(f"{group}.{key}", None)
    if val is None:
        errors.append(f"{group}.{key} is required")
        return None
    if typ is not None and not isinstance(val, typ):
        errors.append(f"{group}.{key}={val!r} must be {typ}")
    return val


def _choice(cfg: Any, group: str, key: str, choices, errors: List[str],
            required=True):
    val = cfg.select(f"{group}.{key}", None)
    if val is None:
        if required:
            errors.append(f"{group}.{key} is required (one of {choices})")
        return None
    if val not in choices:
        errors.append(f"{group}.{key}={val!r} not in {choices}")
    return val


def validate_config(cfg: Any) -> None:
    """Raise ConfigError listing every problem in the composed config."""
    errors: List[str] = []

    # dataset_params
    _choice(cfg, "dataset_params", "dataset_name", DATASETS, errors)
    _req(cfg, "dataset_params", "data_root_dir", errors, str)
    bs = _req(cfg, "dataset_params", "total_batch_size", errors, int)
    if isinstance(bs, int) and bs <= 0:
        errors.append("dataset_params.total_batch_size must be > 0")
    _choice(cfg, "dataset_params", "dataloader_type", DATALOADER_TYPES, errors)

    # model_params
    _req(cfg, "model_params", "model_name", errors, str)
    _choice(cfg, "model_params", "mask_layer_type", MASK_LAYER_TYPES, errors)

    # pruning_params
    method = _choice(cfg, "pruning_params", "prune_method", PRUNE_METHODS, errors)
    ttype = _choice(cfg, "pruning_params", "training_type", TRAINING_TYPES, errors)
    ts = cfg.select("pruning_params.target_sparsity", None)
    if method != "just dont":
        if ts is None:
            errors.append("pruning_params.target_sparsity is required")
        elif not (0.0 <= float(ts) < 1.0):
            errors.append(f"pruning_params.target_sparsity={ts} must be in [0, 1)")
    if method in ITERATIVE_METHODS:
        pr = cfg.select("pruning_params.prune_rate", None)
        if pr is None:
            errors.append(f"pruning_params.prune_rate is required for {method}")
        elif not (0.0 < float(pr) < 1.0):
            errors.append(f"pruning_params.prune_rate={pr} must be in (0, 1)")
    if ttype == "wr" and cfg.select("pruning_params.rewind_epoch", None) is None:
        errors.append("pruning_params.rewind_epoch is required for training_type=wr")

    # experiment_params
    _req(cfg, "experiment_params", "seed", errors, int)
    _req(cfg, "experiment_params", "base_dir", errors, str)
    epl = _req(cfg, "experiment_params", "epochs_per_level", errors, int)
    if isinstance(epl, int) and epl < 0:
        errors.append("experiment_params.epochs_per_level must be >= 0")
    _choice(cfg, "experiment_params", "training_precision", PRECISIONS, errors)

    # optimizer_params
    _req(cfg, "optimizer_params", "lr", errors, (int, float))
    _req(cfg, "optimizer_params", "momentum", errors, (int, float))
    _req(cfg, "optimizer_params", "weight_decay", errors, (int, float))
    sched = _choice(cfg, "optimizer_params", "scheduler_type", SCHEDULER_TYPES, errors)
    if sched == "TrapezoidalSchedule":
        if cfg.select("optimizer_params.trapezoidal_scheduler_stuff", None) is None:
            errors.append("optimizer_params.trapezoidal_scheduler_stuff "
                          "{warmup_steps, cooldown_steps} required for "
                          "TrapezoidalSchedule")

    # cyclic_training (optional group; defaults are fine)
    nc = cfg.select("cyclic_training.num_cycles", 1)
    if not isinstance(nc, int) or nc < 1:
        errors.append(f"cyclic_training.num_cycles={nc!r} must be an int >= 1")
    strat = cfg.select("cyclic_training.strategy", "constant")
    if strat not in CYCLIC_STRATEGIES:
        errors.append(f"cyclic_training.strategy={strat!r} not in {CYCLIC_STRATEGIES}")

    if errors:
        raise ConfigError("invalid config:\n  - " + "\n  - ".join(errors))
