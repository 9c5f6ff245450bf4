"""End-to-end integration: BASELINE config 1 (CIFAR10 ResNet18 ERK
one-shot, 1 epoch, world_size=1, CPU) + IMP multi-level + cyclic +
resume."""

import os

import pandas as pd
import pytest
import torch

from run_experiment import run
from turboprune_amd.config import compose
from turboprune_amd.harness import CyclicPruningHarness


def _base_overrides(tmp_path, extra=()):
    return [
        "experiment_params.epochs_per_level=1",
        "dataset_params.total_batch_size=64",
        "+dataset_params.synthetic_size=256",
        f"experiment_params.base_dir={tmp_path}/experiments",
        f"dataset_params.data_root_dir={tmp_path}/data",
        *extra,
    ]


def test_minimum_slice_erk(tmp_path):
    cfg = compose("cifar10_er_erk", _base_overrides(
        tmp_path, ["pruning_params.target_sparsity=0.9"]))
    expt_dir = run(cfg)

    assert os.path.exists(os.path.join(expt_dir, "expt_config.yaml"))
    assert os.path.exists(os.path.join(expt_dir, "checkpoints",
                                       "model_init.pt"))
    assert os.path.exists(os.path.join(expt_dir, "checkpoints",
                                       "model_level_0.pt"))
    assert os.path.exists(os.path.join(expt_dir, "artifacts",
                                       "optimizer_init.pt"))

    metrics = pd.read_csv(os.path.join(
        expt_dir, "metrics", "level_wise_metrics", "level_0_metrics.csv"))
    for col in ("epoch", "train_loss", "train_acc", "test_loss",
                "test_acc", "max_test_acc", "sparsity"):
        assert col in metrics.columns
    assert len(metrics) == 1
    assert 80.0 < metrics["sparsity"][0] < 100.0

    summaries = [f for f in os.listdir(expt_dir)
                 if f.endswith("_summary.csv")]
    assert len(summaries) == 1
    summary = pd.read_csv(os.path.join(expt_dir, summaries[0]))
    assert list(summary.columns) == ["Level", "Sparsity", "Last_Test_Acc",
                                     "Max_Test_Acc"]

    sd = torch.load(os.path.join(expt_dir, "checkpoints", "model_level_0.pt"),
                    weights_only=True)
    masks = {k: v for k, v in sd.items() if k.endswith("mask")}
    assert len(masks) == 21  # resnet18: 20 convs + fc
    total = sum(v.numel() for v in masks.values())
    zeros = sum(int((v == 0).sum()) for v in masks.values())
    assert zeros / total == pytest.approx(0.9, abs=0.05)


def test_imp_two_levels_rewind(tmp_path):
    cfg = compose("cifar10_er_erk", _base_overrides(tmp_path, [
        "pruning_params=iterative_imp",
        "pruning_params.target_sparsity=0.3",  # ladder [1.0, 0.8, 0.64] → 3 levels
    ]))
    expt_dir = run(cfg)
    ckpts = sorted(os.listdir(os.path.join(expt_dir, "checkpoints")))
    assert "model_level_0.pt" in ckpts
    assert "model_level_1.pt" in ckpts
    assert "model_level_2.pt" in ckpts

    # sparsity grows level over level
    def sparsity(name):
        sd = torch.load(os.path.join(expt_dir, "checkpoints", name),
                        weights_only=True)
        masks = [v for k, v in sd.items() if k.endswith("mask")]
        return sum(int((v == 0).sum()) for v in masks) / \
            sum(v.numel() for v in masks)

    s0, s1, s2 = (sparsity(f"model_level_{i}.pt") for i in range(3))
    assert s0 == 0.0
    assert s1 == pytest.approx(0.2, abs=0.02)
    assert s2 == pytest.approx(0.36, abs=0.02)  # 1 - 0.8^2

    # IMP: non-mask weights of level-2 checkpoint start = init (rewound);
    # level ckpts are post-training so just check summary has 3 rows
    summaries = [f for f in os.listdir(expt_dir) if f.endswith("_summary.csv")]
    assert len(pd.read_csv(os.path.join(expt_dir, summaries[0]))) == 3


def test_wr_rewind_checkpoint_written(tmp_path):
    cfg = compose("cifar10_er_erk", _base_overrides(tmp_path, [
        "pruning_params=iterative_wr",
        "pruning_params.target_sparsity=0.2",
        "pruning_params.rewind_epoch=0",
        "experiment_params.epochs_per_level=2",
    ]))
    expt_dir = run(cfg)
    assert os.path.exists(os.path.join(expt_dir, "checkpoints",
                                       "model_rewind.pt"))
    assert os.path.exists(os.path.join(expt_dir, "artifacts",
                                       "optimizer_rewind.pt"))


def test_cyclic_harness(tmp_path):
    cfg = compose("cifar10_er_erk", _base_overrides(tmp_path, [
        "pruning_params.target_sparsity=0.5",
        "experiment_params.epochs_per_level=2",
    ]))
    cfg.cyclic_training.num_cycles = 2
    cfg.cyclic_training.strategy = "constant"
    expt_dir = run(cfg, harness_cls=CyclicPruningHarness)
    metrics = pd.read_csv(os.path.join(
        expt_dir, "metrics", "level_wise_metrics", "level_0_metrics.csv"))
    assert "cycle" in metrics.columns
    assert sorted(metrics["cycle"].unique().tolist()) == [0, 1]
    summaries = [f for f in os.listdir(expt_dir) if f.endswith("_summary.csv")]
    summary = pd.read_csv(os.path.join(expt_dir, summaries[0]))
    assert "Schedule" in summary.columns


def test_resume_from_level(tmp_path):
    cfg = compose("cifar10_er_erk", _base_overrides(tmp_path, [
        "pruning_params=iterative_imp",
        "pruning_params.target_sparsity=0.3",
    ]))
    expt_dir = run(cfg)
    name = os.path.basename(expt_dir)
    # resume at level 2 of the finished experiment
    cfg2 = compose("cifar10_er_erk", _base_overrides(tmp_path, [
        "pruning_params=iterative_imp",
        "pruning_params.target_sparsity=0.3",
        "experiment_params.resume_experiment=true",
        "+experiment_params.resume_experiment_stuff.resume_level=2",
        f"+experiment_params.resume_experiment_stuff.resume_expt_name={name}",
    ]))
    expt_dir2 = run(cfg2)
    assert expt_dir2 == expt_dir
    assert os.path.exists(os.path.join(expt_dir, "checkpoints",
                                       "model_level_2.pt"))


def test_snip_end_to_end(tmp_path):
    """SNIP prune-at-init (data-driven scorer) through the full driver."""
    cfg = compose("cifar10_er_snip", _base_overrides(
        tmp_path, ["pruning_params.target_sparsity=0.8"]))
    expt_dir = run(cfg)
    sd = torch.load(os.path.join(expt_dir, "checkpoints",
                                 "model_level_0.pt"), weights_only=True)
    masks = [v for k, v in sd.items() if k.endswith("mask")]
    total = sum(v.numel() for v in masks)
    zeros = sum(int((v == 0).sum()) for v in masks)
    assert zeros / total == pytest.approx(0.8, abs=0.02)


def test_synflow_end_to_end(tmp_path):
    cfg = compose("cifar10_er_synflow", _base_overrides(
        tmp_path, ["pruning_params.target_sparsity=0.8"]))
    expt_dir = run(cfg)
    sd = torch.load(os.path.join(expt_dir, "checkpoints",
                                 "model_level_0.pt"), weights_only=True)
    masks = [v for k, v in sd.items() if k.endswith("mask")]
    total = sum(v.numel() for v in masks)
    zeros = sum(int((v == 0).sum()) for v in masks)
    assert zeros / total == pytest.approx(0.8, abs=0.02)


def test_schedule_free_harness(tmp_path):
    cfg = compose("cifar10_er_erk", _base_overrides(tmp_path, [
        "pruning_params.target_sparsity=0.5",
        "optimizer_params.scheduler_type=ScheduleFree",
    ]))
    expt_dir = run(cfg)
    summaries = [f for f in os.listdir(expt_dir) if f.endswith("_summary.csv")]
    assert len(summaries) == 1


def test_imagenet_synthetic_end_to_end(tmp_path):
    """ImageNet-shaped synthetic run through the driver (world_size=1,
    CPU): exercises the ImageNet loader + harness path."""
    cfg = compose("bench_resnet50_imagenet", [
        "model_params.model_name=resnet18",
        "experiment_params.epochs_per_level=1",
        "experiment_params.distributed=false",
        "dataset_params.total_batch_size=8",
        "+dataset_params.steps_per_epoch=2",
        f"experiment_params.base_dir={tmp_path}/experiments",
        f"dataset_params.data_root_dir={tmp_path}/data",
        "pruning_params=pai_er_erk",
        "pruning_params.target_sparsity=0.5",
    ])
    expt_dir = run(cfg)
    assert os.path.exists(os.path.join(expt_dir, "checkpoints",
                                       "model_level_0.pt"))


def test_cyclic_driver_main(tmp_path, monkeypatch):
    """run_cyclic_training_experiment.main end-to-end on CPU (the
    reference's cyclic driver crashes as shipped)."""
    import os

    from run_cyclic_training_experiment import main as cyclic_main
    monkeypatch.setenv("TURBOPRUNE_SYNTHETIC_CIFAR", "1")
    expt_dir = cyclic_main([
        "--config-name=cifar10_er_erk",
        f"experiment_params.base_dir={tmp_path}",
        "experiment_params.epochs_per_level=2",
        "dataset_params.total_batch_size=32",
        "+dataset_params.synthetic_size=64",
        "cyclic_training.num_cycles=2",
        "cyclic_training.strategy=constant",
        "pruning_params.target_sparsity=0.5",
    ])
    assert os.path.isdir(expt_dir)
    level0 = os.path.join(expt_dir, "metrics", "level_wise_metrics",
                          "level_0_metrics.csv")
    assert os.path.exists(level0)
    with open(level0) as f:
        header = f.readline()
    assert "cycle" in header
