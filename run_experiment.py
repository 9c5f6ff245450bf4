#!/usr/bin/env python3
"""Experiment driver: lottery-ticket pruning across sparsity levels.

Usage (mirrors the reference's hydra CLI, run_experiment.py:21):

    python run_experiment.py --config-name=cifar10_er_erk \
        pruning_params.target_sparsity=0.9 experiment_params.epochs_per_level=2

Distributed (ImageNet only, one process per GPU over RCCL):

    torchrun --nproc_per_node=8 --master-addr 127.0.0.1 run_experiment.py \
        --config-name=imagenet_imp

Per-level flow (reference: run_experiment.py:84-126): rank 0 prunes /
rewinds the masked model, state is broadcast to all ranks, a fresh
optimizer+schedule trains the level, and ``model_level_{L}.pt`` is saved.
"""

from __future__ import annotations

import os
import sys

import torch
import torch.distributed as dist

from turboprune_amd.config import compose, validate_config
from turboprune_amd.config.compose import parse_cli, save_config
from turboprune_amd.harness import CyclicPruningHarness, PruningHarness
from turboprune_amd.parallel import (broadcast_object,
                                     broadcast_model_state,
                                     cleanup_distributed, setup_distributed)
from turboprune_amd.pruning import prune_the_model
from turboprune_amd.utils.experiment import (gen_expt_dir,
                                             generate_densities,
                                             resume_experiment, save_model,
                                             set_seed)
from turboprune_amd.utils.logging import WandbShim


def run(cfg, harness_cls=PruningHarness):
    validate_config(cfg)

    env_world = int(os.environ.get("WORLD_SIZE", 1))
    is_cifar = cfg.dataset_params.dataset_name in ("CIFAR10", "CIFAR100")
    if is_cifar and env_world > 1:
        # CIFAR x torchrun guard (reference: run_experiment.py:25-37)
        print("CIFAR runs are single-GPU; do not launch with torchrun.")
        sys.exit(1)

    set_seed(cfg)
    rank, local_rank, world_size = 0, 0, 1
    if env_world > 1:
        rank, local_rank, world_size = setup_distributed()

    resume_level = 0
    if rank == 0:
        if cfg.select("experiment_params.resume_experiment", False):
            prefix, expt_dir, resume_level = resume_experiment(cfg)
        else:
            prefix, expt_dir = gen_expt_dir(cfg)
            save_config(expt_dir, cfg)
    else:
        prefix, expt_dir = "", ""
    if world_size > 1:
        prefix, expt_dir, resume_level = broadcast_object(
            (prefix, expt_dir, resume_level))

    wandb_run = WandbShim(
        enabled=bool(os.environ.get("TURBOPRUNE_WANDB", "")) and rank == 0,
        project=cfg.select("experiment_params.wandb_project_name"),
        name=prefix, config=cfg.to_dict())

    harness = harness_cls(cfg, gpu_id=local_rank, expt_dir=expt_dir,
                          prefix=prefix)
    model = harness.model  # PruneModel
    inner = model.model

    densities = generate_densities(cfg, current_sparsity=0.0)
    at_init = cfg.pruning_params.training_type == "at_init"
    iterative = cfg.pruning_params.training_type in ("imp", "wr", "lrr")
    epochs_per_level = int(cfg.experiment_params.epochs_per_level)

    for level, density in enumerate(densities):
        if level < resume_level:
            continue
        if harness.is_rank0:
            if level == resume_level and resume_level > 0 and iterative:
                model.load_model(os.path.join(
                    expt_dir, "checkpoints", f"model_level_{level - 1}.pt"))
            if level == 0 and at_init:
                prune_the_model(cfg, inner, density,
                                dataloader=harness.train_loader,
                                device=harness.device)
            elif level > 0 and iterative:
                prune_the_model(cfg, inner, density,
                                dataloader=harness.train_loader,
                                device=harness.device)
                model.reset_weights(cfg, expt_dir)
            print(f"[level {level}] density {density:.5f} "
                  f"overall sparsity {model.get_overall_sparsity():.2f}%")
        if world_size > 1:
            # propagate rank-0 pruning/rewinding (C4 equivalent)
            broadcast_model_state(inner)
            model.refresh_caches()
        elif harness.is_rank0:
            model.refresh_caches()

        result = harness.train_one_level(epochs_per_level, level)
        if harness.is_rank0:
            save_model(model, os.path.join(
                expt_dir, "checkpoints", f"model_level_{level}.pt"),
                harness.distributed)
            wandb_run.log({"level": level,
                           "sparsity": result["sparsity"],
                           "max_test_acc": result["max_test_acc"]})
        if world_size > 1:
            dist.barrier()

    wandb_run.finish()
    cleanup_distributed()
    return expt_dir


def main(argv=None):
    config_name, overrides = parse_cli(argv if argv is not None
                                       else sys.argv[1:])
    cfg = compose(config_name, overrides)
    return run(cfg, PruningHarness)


if __name__ == "__main__":
    main()
