"""Distributed backend: torch.distributed over RCCL / xGMI.

Single node, one process per GPU. The ``"nccl"`` backend string IS RCCL
on ROCm builds (reference call: distributed_utils.py:63-66); CPU tests
use gloo.

MI355X-specific choices:

- ``broadcast_buffers=False`` on DDP: masks are buffers and immutable
  between prune events — the reference re-broadcasts the full mask set
  every forward (C3, weights-sized payload per step); here masks/weights
  are broadcast explicitly once per level / prune event
  (``broadcast_model_state``), identical observable behavior.
- DDP bucket size defaults to 32 MB. Two opposing pressures: xGMI is 7
  point-to-point links per GPU (~153 GB/s each) and ring all-reduce is
  per-link bound, favoring large messages — but ResNet50's gradients
  total only ~102 MB, so a 128 MB cap degenerates to ONE bucket and the
  all-reduce runs fully exposed after backward. 32 MB gives ~4 buckets
  (4 MB per ring step per link — still link-efficient) and lets the
  collectives overlap backward. Override with cfg
  ``experiment_params.bucket_cap_mb``.
- ``gradient_as_bucket_view=True``: no grad copy into buckets.
"""

from __future__ import annotations

import datetime
import hashlib
import os
from typing import Any, Optional, Tuple

import torch
import torch.distributed as dist
from torch.nn.parallel import DistributedDataParallel as DDP

DEFAULT_BUCKET_CAP_MB = 32


def world_info() -> Tuple[int, int, int]:
    """(rank, local_rank, world_size) from env (torchrun) or (0,0,1)."""
    rank = int(os.environ.get("RANK", 0))
    local_rank = int(os.environ.get("LOCAL_RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    return rank, local_rank, world


def setup_distributed(backend: Optional[str] = None,
                      timeout_sec: int = 1800) -> Tuple[int, int, int]:
    """init_process_group + device binding. Returns (rank, local_rank,
    world_size). Backend: nccl(=RCCL) when CUDA/ROCm devices exist,
    else gloo."""
    rank, local_rank, world = world_info()
    if world <= 1 and "MASTER_ADDR" not in os.environ:
        return rank, local_rank, world
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    if not dist.is_initialized():
        dist.init_process_group(backend=backend,
                                timeout=datetime.timedelta(seconds=timeout_sec))
    if torch.cuda.is_available():
        torch.cuda.set_device(local_rank)
    return rank, local_rank, world


def cleanup_distributed() -> None:
    if dist.is_initialized():
        dist.destroy_process_group()


def is_rank0() -> bool:
    return not dist.is_initialized() or dist.get_rank() == 0


def broadcast_object(obj: Any, src: int = 0) -> Any:
    """Broadcast a small picklable object from src (reference:
    distributed_utils.py:7-11)."""
    if not dist.is_initialized():
        return obj
    box = [obj]
    dist.broadcast_object_list(box, src=src)
    return box[0]


@torch.no_grad()
def broadcast_model_state(model: torch.nn.Module, src: int = 0) -> None:
    """Broadcast ALL parameters and buffers (including masks) from src.

    This replaces the reference's implicit DDP-construction sync (C4,
    base_harness.py:81) as the mechanism that propagates rank-0-only
    pruning/rewinding to the other ranks — done once per level/prune
    event instead of per harness rebuild."""
    if not dist.is_initialized():
        return
    tensors = list(model.parameters()) + list(model.buffers())
    for t in tensors:
        dist.broadcast(t.data, src=src)


@torch.no_grad()
def check_model_equality(model: torch.nn.Module) -> bool:
    """Cross-rank parameter hash comparison (working version of the
    reference's dead-code comparator, distributed_utils.py:31-60).
    Used as the DDP-correctness test assertion."""
    if not dist.is_initialized():
        return True
    h = hashlib.sha256()
    for t in list(model.parameters()) + list(model.buffers()):
        h.update(t.detach().float().cpu().numpy().tobytes())
    digest = h.hexdigest()
    digests = [None] * dist.get_world_size()
    dist.all_gather_object(digests, digest)
    return all(d == digests[0] for d in digests)


def wrap_ddp(model: torch.nn.Module, cfg: Any = None,
             device: Optional[torch.device] = None) -> torch.nn.Module:
    """DDP wrap tuned for xGMI. No-op when not distributed."""
    if not dist.is_initialized() or dist.get_world_size() == 1:
        return model
    bucket_mb = DEFAULT_BUCKET_CAP_MB
    if cfg is not None:
        bucket_mb = int(cfg.select("experiment_params.bucket_cap_mb",
                                   DEFAULT_BUCKET_CAP_MB))
    kwargs = dict(broadcast_buffers=False,
                  bucket_cap_mb=bucket_mb,
                  gradient_as_bucket_view=True)
    if device is not None and device.type == "cuda":
        kwargs["device_ids"] = [device.index]
    # explicit initial sync (masks included) — C4 equivalent
    broadcast_model_state(model, src=0)
    return DDP(model, **kwargs)
