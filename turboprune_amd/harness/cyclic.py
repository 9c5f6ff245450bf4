"""CyclicPruningHarness: cycle loop inside each level.

Working version of the reference's cyclic harness (harness_definitions/
cyclic_harness.py — crashes as shipped): per-cycle epoch
budget from ``generate_cyclical_schedule``, fresh optimizer+scheduler per
cycle, rewind checkpoint at level 0 / cycle 0, cycle column in metrics
and a Schedule column in the summary.
"""

from __future__ import annotations


from typing import Any, Optional

import torch

from turboprune_amd.harness.pruning import PruningHarness
from turboprune_amd.models import PruneModel
from turboprune_amd.optim import build_optimizer, build_scheduler
from turboprune_amd.utils.experiment import (generate_cyclical_schedule,
                                             save_model)


class CyclicPruningHarness(PruningHarness):
    def __init__(self, cfg: Any, gpu_id: int, expt_dir: str,
                 prefix: str = "", model: Optional[PruneModel] = None):
        super().__init__(cfg, gpu_id, expt_dir, prefix, model)
        self.epoch_schedule = generate_cyclical_schedule(cfg)

    def train_one_level(self, epochs_per_level: int, level: int,
                        num_cycles: Optional[int] = None) -> dict:
        num_cycles = num_cycles or int(
            self.cfg.select("cyclic_training.num_cycles", 1))
        schedule = self.epoch_schedule
        rewind_epoch = self.cfg.select("pruning_params.rewind_epoch", None)

        rows = []
        max_test_acc = 0.0
        last_test_acc = 0.0
        total_epochs = 0
        sparsity = self.model.get_overall_sparsity()

        for cycle in range(num_cycles):
            cycle_epochs = schedule[cycle % len(schedule)]
            # fresh optimizer+scheduler per cycle (reference:
            # cyclic_harness.py:193-194)
            self.optimizer = build_optimizer(self.cfg, self.model)
            steps_per_epoch = max(len(self.train_loader), 1)
            self.scheduler = build_scheduler(self.cfg, self.optimizer,
                                             steps_per_epoch, cycle_epochs)

            if level == 0 and cycle == 0 and self.is_rank0:
                save_model(self.model, self._ckpt("model_init.pt"),
                           self.distributed)
                torch.save(self.optimizer.state_dict(),
                           self._artifact("optimizer_init.pt"))

            for epoch in range(cycle_epochs):
                train_loss, train_acc = self.train_epoch(epoch)
                test_loss, test_acc = self.test()
                max_test_acc = max(max_test_acc, test_acc)
                last_test_acc = test_acc
                if (level == 0 and cycle == 0 and rewind_epoch is not None
                        and epoch == int(rewind_epoch) and self.is_rank0):
                    save_model(self.model, self._ckpt("model_rewind.pt"),
                               self.distributed)
                    torch.save(self.optimizer.state_dict(),
                               self._artifact("optimizer_rewind.pt"))
                rows.append({
                    "cycle": cycle,
                    "epoch": total_epochs + epoch,
                    "train_loss": round(train_loss, 6),
                    "train_acc": round(train_acc, 4),
                    "test_loss": round(test_loss, 6),
                    "test_acc": round(test_acc, 4),
                    "max_test_acc": round(max_test_acc, 4),
                    "sparsity": round(sparsity, 4),
                })
            total_epochs += cycle_epochs

        if self.is_rank0:
            self.logger.write_level_csv(level, rows)
            self.logger.append_summary({
                "Level": level,
                "Sparsity": round(sparsity, 4),
                "Last_Test_Acc": round(last_test_acc, 4),
                "Max_Test_Acc": round(max_test_acc, 4),
                "Schedule": "-".join(map(str, schedule)),
            })
        return {"train_rows": rows, "max_test_acc": max_test_acc,
                "last_test_acc": last_test_acc, "sparsity": sparsity}
