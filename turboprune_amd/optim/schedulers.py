"""LR schedules (reference: utils/schedulers.py).

The reference ships four working schedules (TriangularSchedule,
TrapezoidalSchedule, OneCycleLR, ScheduleFree) and two declared-but-broken
ones (MultiStepLRWarmup / ImageNetLRDropsWarmup — SURVEY §2.6.4); all six
work here. Each scheduler carries ``step_granularity`` ("step" or
"epoch") so the harness knows when to step it (reference:
base_harness.py:178-189).
"""

from __future__ import annotations

from typing import Any, Optional

import numpy as np
import torch
from torch.optim import Optimizer

STEP_WISE = ("TriangularSchedule", "TrapezoidalSchedule", "OneCycleLR")
EPOCH_WISE = ("MultiStepLRWarmup", "ImageNetLRDropsWarmup")


def TriangularSchedule(cfg: Any, optimizer: Optimizer, steps_per_epoch: int,
                       epochs_per_level: Optional[int] = None):
    """Per-step triangular LR: np.interp over
    [0, warmup_fraction*T, T] -> [0.2, 1, 0]
    (reference: schedulers.py:79-117)."""
    epochs = int(epochs_per_level if epochs_per_level is not None
                 else cfg.experiment_params.epochs_per_level)
    total = epochs * steps_per_epoch
    warmup = float(cfg.optimizer_params.warmup_fraction)
    lr_schedule = np.interp(np.arange(1 + total),
                            [0, int(warmup * total), total], [0.2, 1, 0])
    sched = torch.optim.lr_scheduler.LambdaLR(optimizer,
                                              lr_schedule.__getitem__)
    sched.step_granularity = "step"
    return sched


def _step_trapezoidal(it, num_iterations, warmup_iters, warmdown_iters):
    assert it <= num_iterations
    if it < warmup_iters:
        return (it + 1) / warmup_iters
    if it < num_iterations - warmdown_iters:
        return 1.0
    return (num_iterations - it) / warmdown_iters


def TrapezoidalSchedule(cfg: Any, optimizer: Optimizer, steps_per_epoch: int,
                        epochs_per_level: Optional[int] = None):
    """Warmup -> flat -> warmdown per-step schedule
    (reference: schedulers.py:65-76,120-143)."""
    epochs = int(epochs_per_level if epochs_per_level is not None
                 else cfg.experiment_params.epochs_per_level)
    total = epochs * steps_per_epoch
    stuff = cfg.select("optimizer_params.trapezoidal_scheduler_stuff")
    warmup = int(stuff["warmup_steps"])
    cooldown = int(stuff["cooldown_steps"])
    table = [_step_trapezoidal(it, total, warmup, cooldown)
             for it in range(1 + total)]
    sched = torch.optim.lr_scheduler.LambdaLR(optimizer, table.__getitem__)
    sched.step_granularity = "step"
    return sched


def OneCycleLR(cfg: Any, optimizer: Optimizer, steps_per_epoch: int,
               epochs_per_level: Optional[int] = None):
    epochs = int(epochs_per_level if epochs_per_level is not None
                 else cfg.experiment_params.epochs_per_level)
    sched = torch.optim.lr_scheduler.OneCycleLR(
        optimizer, max_lr=float(cfg.optimizer_params.lr),
        epochs=max(epochs, 1), steps_per_epoch=max(steps_per_epoch, 1))
    sched.step_granularity = "step"
    return sched


def MultiStepLRWarmup(cfg: Any, optimizer: Optimizer,
                      steps_per_epoch: int = 0,
                      epochs_per_level: Optional[int] = None):
    """Linear warmup (0.1 -> 1.0 over warmup_fraction*epochs) then
    MultiStepLR drops at epochs [60, 120] — per-epoch stepping
    (working version of reference schedulers.py:8-34)."""
    epochs = int(epochs_per_level if epochs_per_level is not None
                 else cfg.experiment_params.epochs_per_level)
    warmup_epochs = max(int(float(cfg.optimizer_params.warmup_fraction)
                            * epochs), 1)
    warm = torch.optim.lr_scheduler.LinearLR(
        optimizer, start_factor=0.1, end_factor=1.0,
        total_iters=warmup_epochs)
    main = torch.optim.lr_scheduler.MultiStepLR(
        optimizer, milestones=[60, 120], gamma=0.1)
    sched = torch.optim.lr_scheduler.SequentialLR(
        optimizer, schedulers=[warm, main], milestones=[warmup_epochs])
    sched.step_granularity = "epoch"
    return sched


def ImageNetLRDropsWarmup(cfg: Any, optimizer: Optimizer,
                          steps_per_epoch: int = 0,
                          epochs_per_level: Optional[int] = None):
    """10-epoch linear warmup then drops at epochs [40, 70] — per-epoch
    stepping (working version of reference schedulers.py:37-62)."""
    warm = torch.optim.lr_scheduler.LinearLR(
        optimizer, start_factor=0.1, end_factor=1.0, total_iters=10)
    main = torch.optim.lr_scheduler.MultiStepLR(
        optimizer, milestones=[40, 70], gamma=0.1)
    sched = torch.optim.lr_scheduler.SequentialLR(
        optimizer, schedulers=[warm, main], milestones=[10])
    sched.step_granularity = "epoch"
    return sched


def build_scheduler(cfg: Any, optimizer: Optimizer, steps_per_epoch: int,
                    epochs_per_level: Optional[int] = None):
    """Scheduler dispatch (reference: standard_pruning_harness.py:86-119).
    Returns None for ScheduleFree (the optimizer schedules itself)."""
    stype = cfg.optimizer_params.scheduler_type
    if stype == "ScheduleFree":
        return None
    fn = globals().get(stype)
    if fn is None:
        raise ValueError(f"unknown scheduler_type {stype!r}")
    return fn(cfg, optimizer, steps_per_epoch, epochs_per_level)
