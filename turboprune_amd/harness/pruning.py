"""PruningHarness: per-level training with checkpoints/CSVs.

Reference: harness_definitions/standard_pruning_harness.py. Artifact
formats match exactly:
- ``checkpoints/model_init.pt`` (level 0 start), ``model_rewind.pt``
  (at pruning_params.rewind_epoch of level 0), ``model_level_{L}.pt``;
- ``artifacts/optimizer_init.pt`` / ``optimizer_rewind.pt``;
- ``metrics/level_wise_metrics/level_{L}_metrics.csv`` with columns
  epoch / train_loss / train_acc / test_loss / test_acc / max_test_acc /
  sparsity;
- ``{prefix}_summary.csv`` appended per level (Level, Sparsity,
  Last_Test_Acc, Max_Test_Acc).
"""

from __future__ import annotations

import os
from typing import Any, Optional

import torch

from turboprune_amd.harness.base import BaseHarness
from turboprune_amd.models import PruneModel
from turboprune_amd.optim import build_optimizer, build_scheduler
from turboprune_amd.utils.experiment import save_model
from turboprune_amd.utils.logging import MetricsLogger


class PruningHarness(BaseHarness):
    def __init__(self, cfg: Any, gpu_id: int, expt_dir: str,
                 prefix: str = "", model: Optional[PruneModel] = None):
        super().__init__(cfg, gpu_id, expt_dir, prefix, model)
        self.logger = MetricsLogger(expt_dir, prefix, self.is_rank0)

    # ------------------------------------------------------------------
    def setup_level(self, epochs_per_level: int) -> None:
        """Fresh optimizer + scheduler per level (reference semantics:
        new harness per level, run_experiment.py:113-115)."""
        self.optimizer = build_optimizer(self.cfg, self.model)
        steps_per_epoch = max(len(self.train_loader), 1)
        self.scheduler = build_scheduler(self.cfg, self.optimizer,
                                         steps_per_epoch, epochs_per_level)
        self.lr_history = []

    def _ckpt(self, name: str) -> str:
        return os.path.join(self.expt_dir, "checkpoints", name)

    def _artifact(self, name: str) -> str:
        return os.path.join(self.expt_dir, "artifacts", name)

    # ------------------------------------------------------------------
    def train_one_level(self, epochs_per_level: int, level: int) -> dict:
        self.setup_level(epochs_per_level)
        rewind_epoch = self.cfg.select("pruning_params.rewind_epoch", None)

        if level == 0 and self.is_rank0:
            save_model(self.model, self._ckpt("model_init.pt"),
                       self.distributed)
            torch.save(self.optimizer.state_dict(),
                       self._artifact("optimizer_init.pt"))

        rows = []
        max_test_acc = 0.0
        last_test_acc = 0.0
        sparsity = self.model.get_overall_sparsity() \
            if hasattr(self.model, "get_overall_sparsity") else 0.0
        for epoch in range(epochs_per_level):
            train_loss, train_acc = self.train_epoch(epoch)
            test_loss, test_acc = self.test()
            max_test_acc = max(max_test_acc, test_acc)
            last_test_acc = test_acc

            if (level == 0 and rewind_epoch is not None
                    and epoch == int(rewind_epoch) and self.is_rank0):
                save_model(self.model, self._ckpt("model_rewind.pt"),
                           self.distributed)
                torch.save(self.optimizer.state_dict(),
                           self._artifact("optimizer_rewind.pt"))

            rows.append({
                "epoch": epoch,
                "train_loss": round(train_loss, 6),
                "train_acc": round(train_acc, 4),
                "test_loss": round(test_loss, 6),
                "test_acc": round(test_acc, 4),
                "max_test_acc": round(max_test_acc, 4),
                "sparsity": round(sparsity, 4),
            })
            if self.is_rank0:
                self.logger.log({
                    "level": level, "epoch": epoch,
                    "train_loss": train_loss, "train_acc": train_acc,
                    "test_loss": test_loss, "test_acc": test_acc,
                    "images_per_sec":
                        self.throughput.images_per_sec * self.world_size,
                    "ms_per_step": self.throughput.ms_per_step,
                    "lr": self.lr_history[-1] if self.lr_history else None,
                    "sparsity": sparsity,
                })

        if self.is_rank0:
            self.logger.write_level_csv(level, rows)
            self.logger.append_summary({
                "Level": level,
                "Sparsity": round(sparsity, 4),
                "Last_Test_Acc": round(last_test_acc, 4),
                "Max_Test_Acc": round(max_test_acc, 4),
            })
        return {"train_rows": rows, "max_test_acc": max_test_acc,
                "last_test_acc": last_test_acc, "sparsity": sparsity}
