import pytest
import torch
import torch.nn as nn

from turboprune_amd.config import compose
from turboprune_amd.optim.schedulers import build_scheduler


def _opt(lr=0.2):
    return torch.optim.SGD(nn.Linear(2, 2).parameters(), lr=lr)


def test_triangular_shape():
    cfg = compose("cifar10_er_erk", ["experiment_params.epochs_per_level=10"])
    opt = _opt(0.2)
    sched = build_scheduler(cfg, opt, steps_per_epoch=10)
    assert sched.step_granularity == "step"
    lrs = [opt.param_groups[0]["lr"]]
    for _ in range(100):
        sched.step()
        lrs.append(opt.param_groups[0]["lr"])
    # starts at 0.2*base, peaks at warmup_fraction (20), ends at 0
    assert lrs[0] == pytest.approx(0.2 * 0.2)
    assert max(lrs) == pytest.approx(0.2)
    assert lrs[20] == pytest.approx(0.2)
    assert lrs[-1] == pytest.approx(0.0, abs=1e-9)


def test_trapezoidal_shape():
    cfg = compose("cifar10_er_erk", [
        "experiment_params.epochs_per_level=10",
        "optimizer_params.scheduler_type=TrapezoidalSchedule",
        "+optimizer_params.trapezoidal_scheduler_stuff.warmup_steps=10",
        "+optimizer_params.trapezoidal_scheduler_stuff.cooldown_steps=20",
    ])
    opt = _opt(0.2)
    sched = build_scheduler(cfg, opt, steps_per_epoch=10)
    lrs = [opt.param_groups[0]["lr"]]
    for _ in range(100):
        sched.step()
        lrs.append(opt.param_groups[0]["lr"])
    assert lrs[50] == pytest.approx(0.2)  # plateau
    assert lrs[-1] == pytest.approx(0.0, abs=1e-9)
    assert lrs[0] == pytest.approx(0.02)  # (0+1)/10 * 0.2


def test_multistep_warmup_epochwise():
    cfg = compose("cifar10_er_erk", [
        "experiment_params.epochs_per_level=150",
        "optimizer_params.scheduler_type=MultiStepLRWarmup"])
    opt = _opt(0.2)
    sched = build_scheduler(cfg, opt, steps_per_epoch=100)
    assert sched.step_granularity == "epoch"
    lrs = []
    for _ in range(150):
        lrs.append(opt.param_groups[0]["lr"])
        sched.step()
    # SequentialLR semantics (as in the reference, schedulers.py:8-34):
    # milestones [60,120] count from the end of the 30-epoch warmup
    assert lrs[0] == pytest.approx(0.02)      # 0.1 * base warmup start
    assert lrs[30] == pytest.approx(0.2)      # warmed up
    assert lrs[89] == pytest.approx(0.2)      # full lr before the drop
    assert lrs[91] == pytest.approx(0.02)     # dropped 10x at 30+60


def test_imagenet_drops_warmup():
    cfg = compose("imagenet_imp", [
        "optimizer_params.scheduler_type=ImageNetLRDropsWarmup"])
    opt = _opt(0.2)
    sched = build_scheduler(cfg, opt, steps_per_epoch=100)
    assert sched.step_granularity == "epoch"
    lrs = []
    for _ in range(90):
        lrs.append(opt.param_groups[0]["lr"])
        sched.step()
    # drops at global epoch 10+40=50 and 10+70=80 (SequentialLR, as in
    # the reference's ImageNetLRDropsWarmup)
    assert lrs[0] == pytest.approx(0.02)
    assert lrs[15] == pytest.approx(0.2)
    assert lrs[49] == pytest.approx(0.2)
    assert lrs[51] == pytest.approx(0.02)
    assert lrs[81] == pytest.approx(0.002)


def test_onecycle_builds():
    cfg = compose("cifar10_er_erk", [
        "experiment_params.epochs_per_level=5",
        "optimizer_params.scheduler_type=OneCycleLR"])
    opt = _opt(0.2)
    sched = build_scheduler(cfg, opt, steps_per_epoch=10)
    assert sched.step_granularity == "step"
    for _ in range(49):
        sched.step()


def test_schedule_free_returns_none():
    cfg = compose("cifar10_er_erk", [
        "optimizer_params.scheduler_type=ScheduleFree"])
    assert build_scheduler(cfg, _opt(), 10) is None
