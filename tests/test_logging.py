"""Observability utils: JSONL run log, level CSVs, summary append,
throughput meter, wandb shim no-op behavior."""

import csv
import json
import os

from turboprune_amd.utils.logging import (MetricsLogger, Throughput,
                                          WandbShim)


def test_metrics_logger_files(tmp_path):
    d = str(tmp_path)
    os.makedirs(os.path.join(d, "metrics", "level_wise_metrics"))
    lg = MetricsLogger(d, "pfx", is_rank0=True)
    lg.log({"level": 0, "epoch": 0, "train_loss": 1.5})
    lg.write_level_csv(0, [{"epoch": 0, "train_loss": 1.5,
                            "test_acc": 10.0}])
    lg.append_summary({"Level": 0, "Sparsity": 0.0,
                       "Last_Test_Acc": 10.0, "Max_Test_Acc": 10.0})
    lg.append_summary({"Level": 1, "Sparsity": 0.2,
                       "Last_Test_Acc": 12.0, "Max_Test_Acc": 12.0})

    with open(os.path.join(d, "metrics", "run_log.jsonl")) as f:
        rows = [json.loads(l) for l in f]
    assert rows[0]["train_loss"] == 1.5

    lv = os.path.join(d, "metrics", "level_wise_metrics",
                      "level_0_metrics.csv")
    with open(lv) as f:
        r = list(csv.DictReader(f))
    assert r[0]["test_acc"] == "10.0"

    summary = os.path.join(d, "pfx_summary.csv")
    with open(summary) as f:
        r = list(csv.DictReader(f))
    assert len(r) == 2 and r[1]["Level"] == "1"


def test_metrics_logger_nonrank0_writes_nothing(tmp_path):
    d = str(tmp_path)
    os.makedirs(os.path.join(d, "metrics", "level_wise_metrics"))
    lg = MetricsLogger(d, "pfx", is_rank0=False)
    lg.log({"x": 1})
    lg.write_level_csv(0, [{"epoch": 0}])
    lg.append_summary({"Level": 0})
    assert not os.path.exists(os.path.join(d, "metrics", "run_log.jsonl"))
    assert not os.path.exists(os.path.join(d, "pfx_summary.csv"))


def test_throughput_meter():
    t = Throughput()
    t.reset()
    t.start()
    t.step(64)
    t.step(64)
    assert t.images_per_sec >= 0
    assert t.ms_per_step >= 0


def test_wandb_shim_noop():
    run = WandbShim(enabled=False, project=None, name="x", config={})
    run.log({"a": 1})  # must not raise
    run.finish()


def test_display_training_info_smoke(capsys):
    from turboprune_amd.config import compose
    from turboprune_amd.utils.console import display_training_info
    cfg = compose("cifar10_er_erk")
    display_training_info({"dataset": "CIFAR10", "model": "resnet18"},
                          cfg.to_dict())
    out = capsys.readouterr().out
    assert "CIFAR10" in out or "resnet18" in out or len(out) > 0
