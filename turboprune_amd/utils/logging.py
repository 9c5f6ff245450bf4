"""Metrics & observability: CSV writers, JSONL run log, throughput meter.

The reference logs per-level CSVs (``metrics/level_wise_metrics/
level_{L}_metrics.csv``), an appended ``{prefix}_summary.csv`` and wandb
(reference: standard_pruning_harness.py:243-269). wandb is optional here;
a local JSONL log is always written. Per-step host-synchronizing logging
(reference defect, base_harness.py:129-130) is replaced by buffered
rank-0 per-epoch logging.

`Throughput` is the images/sec meter the BASELINE metric is reported with.
"""

from __future__ import annotations

import csv
import json
import os
import time
from typing import Any, Dict, List, Optional


class MetricsLogger:
    """Rank-0 metrics sink: per-level CSVs + summary CSV + JSONL."""

    def __init__(self, expt_dir: str, prefix: str, is_rank0: bool = True):
        self.expt_dir = expt_dir
        self.prefix = prefix
        self.is_rank0 = is_rank0
        self._jsonl_path = os.path.join(expt_dir, "metrics", "run_log.jsonl")

    # --- per-level epoch metrics -----------------------------------------
    def write_level_csv(self, level: int, rows: List[Dict[str, Any]]) -> None:
        if not self.is_rank0 or not rows:
            return
        path = os.path.join(self.expt_dir, "metrics", "level_wise_metrics",
                            f"level_{level}_metrics.csv")
        with open(path, "w", newline="") as f:
            writer = csv.DictWriter(f, fieldnames=list(rows[0].keys()))
            writer.writeheader()
            writer.writerows(rows)

    # --- per-level summary, appended across levels ------------------------
    def append_summary(self, row: Dict[str, Any]) -> None:
        if not self.is_rank0:
            return
        path = os.path.join(self.expt_dir, f"{self.prefix}_summary.csv")
        exists = os.path.exists(path)
        with open(path, "a", newline="") as f:
            writer = csv.DictWriter(f, fieldnames=list(row.keys()))
            if not exists:
                writer.writeheader()
            writer.writerow(row)

    # --- free-form structured log -----------------------------------------
    def log(self, record: Dict[str, Any]) -> None:
        if not self.is_rank0:
            return
        record = dict(record)
        record.setdefault("time", time.time())
        with open(self._jsonl_path, "a") as f:
            f.write(json.dumps(record) + "\n")


class WandbShim:
    """Optional wandb adapter: real wandb if importable and enabled, no-op
    otherwise (wandb is not installed in the MI355X image)."""

    def __init__(self, enabled: bool, project: Optional[str] = None,
                 name: Optional[str] = None, config: Optional[dict] = None):
        self._run = None
        if enabled:
            try:
                import wandb  # type: ignore
                self._run = wandb.init(project=project, name=name, config=config)
            except Exception:
                self._run = None

    @property
    def run_id(self) -> Optional[str]:
        return getattr(self._run, "id", None)

    def log(self, data: Dict[str, Any], step: Optional[int] = None) -> None:
        if self._run is not None:
            self._run.log(data, step=step)

    def finish(self) -> None:
        if self._run is not None:
            self._run.finish()


class Throughput:
    """Images/sec + step-time meter over a window of steps."""

    def __init__(self):
        self.reset()

    def reset(self) -> None:
        self._t0: Optional[float] = None
        self._images = 0
        self._steps = 0

    def start(self) -> None:
        self._t0 = time.perf_counter()

    def step(self, batch_size: int) -> None:
        if self._t0 is None:
            self.start()
        self._images += batch_size
        self._steps += 1

    @property
    def elapsed(self) -> float:
        return 0.0 if self._t0 is None else time.perf_counter() - self._t0

    @property
    def images_per_sec(self) -> float:
        el = self.elapsed
        return self._images / el if el > 0 else 0.0

    @property
    def ms_per_step(self) -> float:
        return 1000.0 * self.elapsed / self._steps if self._steps else 0.0
