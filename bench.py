#!/usr/bin/env python3
"""Flagship benchmark: masked ResNet50 ImageNet training step
(BASELINE.json headline: "ImageNet epoch time (min) + images/sec,
ResNet50 bs=512 at 1/2/4/8 MI355X").

    python bench.py --gpus N --steps K --warmup W

For N > 1 the driver launches this under torch.distributed.run (one rank
per GPU over RCCL). The global batch stays 512 (the reference's
"effective batch size 512" headline config) so per-GPU work shrinks with
N: strong scaling. Synthetic GPU-resident ImageNet-shaped data (no
network in this environment), random-init weights, bf16 autocast compute
(masked-weight caches in bf16, fused SGD), full train step: data slice +
normalize -> fwd -> CE -> bwd (+DDP all-reduce) -> fused SGD step.

Rank 0 prints ONE JSON line with whole-job images/sec.
"""

from __future__ import annotations

import argparse
import json
import os
import time

import torch
import torch.distributed as dist


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=30)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--global-batch", type=int, default=512)
    p.add_argument("--model", type=str, default="resnet50")
    p.add_argument("--sparsity", type=float, default=0.0,
                   help="optional ER mask sparsity for the masked layers")
    p.add_argument("--dtype", type=str, default="bf16",
                   choices=["bf16", "fp32"])
    p.add_argument("--no-ddp", action="store_true")
    p.add_argument("--graph", action="store_true",
                   help="force hipGraph capture of the train step "
                        "(default: auto-on when per-rank batch <= 128, "
                        "the launch-bound strong-scaling regime)")
    p.add_argument("--no-graph", action="store_true",
                   help="disable hipGraph capture")
    return p.parse_args()


def build_result(args, world, elapsed, loss_val, bf16):
    """The driver-contract JSON line (tests/test_bench_contract.py pins
    the schema): whole-job aggregate value, max-over-ranks elapsed."""
    images = args.steps * args.global_batch
    ips = images / elapsed
    epoch_minutes = 1_281_167 / ips / 60.0
    return {
        "metric": "imagenet_images_per_sec",
        "value": round(ips, 1),
        "unit": "images/s",
        "n_gpus": world,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": round(1000.0 * elapsed / args.steps, 3),
        "higher_is_better": True,
        "scaling": "strong",
        "vs_baseline": None,  # BASELINE.json "published" is empty
        "dtype": "bf16" if bf16 else "fp32",
        "data": "synthetic",
        "config": {
            "model": args.model,
            "global_batch": args.global_batch,
            "seq_len": 224,
            "parallelism": f"dp{world}",
            "sparsity": args.sparsity,
            "epoch_time_min": round(epoch_minutes, 3),
            "final_loss": round(loss_val, 4),
        },
    }


def main():
    args = parse_args()
    rank = int(os.environ.get("RANK", 0))
    local_rank = int(os.environ.get("LOCAL_RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    distributed = world > 1

    assert torch.cuda.is_available(), "bench.py requires a GPU"
    torch.cuda.set_device(local_rank)
    device = torch.device(f"cuda:{local_rank}")
    if distributed:
        dist.init_process_group("nccl")

    from turboprune_amd.config import compose
    from turboprune_amd.data.imagenet import SyntheticImageNet
    from turboprune_amd.models import build_model
    from turboprune_amd.ops import functional as TF
    from turboprune_amd.ops._backend import has_extension
    from turboprune_amd.ops.mask_layers import masked_modules
    from turboprune_amd.optim import FusedMaskedSGD
    from turboprune_amd.parallel.ddp import wrap_ddp

    if not has_extension():
        raise RuntimeError("HIP extension not built; run "
                           "`python setup.py build_ext --inplace` first")

    torch.manual_seed(0)
    bf16 = args.dtype == "bf16"
    per_gpu = args.global_batch // world
    cfg = compose("bench_resnet50_imagenet",
                  [f"model_params.model_name={args.model}"])

    pm = build_model(cfg).to(device).to(memory_format=torch.channels_last)
    if args.sparsity > 0:
        for i, (_, m) in enumerate(masked_modules(pm.model)):
            m.set_er_mask(1.0 - args.sparsity, seed=1234 + i)
    if bf16:
        pm.enable_caches(torch.bfloat16)

    from turboprune_amd.parallel import graph_step
    want_graph = ((args.graph or graph_step.wanted(per_gpu, distributed))
                  and not args.no_graph)

    # graph mode replaces DDP with the captured flat-grad all-reduce;
    # eager mode keeps hook-bucketed DDP (overlap wins at large batch)
    model = pm if want_graph or not (distributed and not args.no_ddp) \
        else wrap_ddp(pm, cfg, device)
    opt = FusedMaskedSGD(pm.parameters(), lr=0.2, momentum=0.9,
                         weight_decay=1e-4, model=pm)

    loader = SyntheticImageNet(per_gpu, device, train=True, image_size=224,
                               pool_size=max(per_gpu * 2, 64),
                               seed=rank,
                               steps_per_epoch=args.steps + args.warmup + 1,
                               dtype=torch.bfloat16 if bf16
                               else torch.float32)

    model.train()

    def one_step(batch):
        x, y = batch
        x = x.to(memory_format=torch.channels_last)
        opt.zero_grad(set_to_none=True)
        with torch.autocast(device_type="cuda", dtype=torch.bfloat16,
                            enabled=bf16):
            out = model(x)
            loss = TF.cross_entropy(out, y)
        loss.backward()
        opt.step()
        return loss

    it = iter(loader)

    if want_graph:
        # captured step (zero -> fwd -> CE -> bwd -> flat all-reduce ->
        # fused SGD): removes ~500 host launches per step; the capture
        # path is the same for 1 and N GPUs (graph_step.py)
        try:
            sx, sy = next(it)
            gstep = graph_step.GraphedTrainStep(
                model, opt, TF.cross_entropy,
                sx.to(memory_format=torch.channels_last), sy, bf16)

            def one_step(batch):
                x, y = batch
                return gstep(x.to(memory_format=torch.channels_last), y)
        except Exception as e:  # noqa: BLE001
            if rank == 0:
                print(f"# hipGraph capture failed ({e!r}); eager fallback",
                      flush=True)
            if distributed and not args.no_ddp:
                model = wrap_ddp(pm, cfg, device)

    for _ in range(args.warmup):
        loss_out = one_step(next(it))

    if distributed:
        dist.barrier()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        loss = one_step(next(it))
    if distributed:
        dist.barrier()
    torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0

    # MAX over ranks
    if distributed:
        t = torch.tensor([elapsed], device=device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    if rank == 0:
        result = build_result(args, world, elapsed, float(loss.item()), bf16)
        print(json.dumps(result))

    if distributed:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
