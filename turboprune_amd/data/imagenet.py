"""ImageNet ingest for MI355X.

Replaces the reference's FFCV pipeline (utils/dataset.py:347-430). Two
modes:

- ``SyntheticImageNet``: a GPU-resident pool of ImageNet-shaped uint8
  images; each step slices a per-rank batch (``total_batch_size //
  world_size``, the reference's sharding math, dataset.py:411) and runs
  the same normalize(+flip) stage the real pipeline uses. This is the
  bench path (BASELINE: synthetic data allowed) and exercises the GPU
  post-decode kernels with zero host traffic. With 288 GB HBM3E per GPU
  a real cached-dataset variant of this loader could hold the entire
  decoded 224² ImageNet train set resident (~193 GB uint8).

- ``ShardedImageNet``: a shard reader for pre-decoded uint8 tensors
  (our beton-equivalent record format: .pt shards of
  {images: uint8 [N,3,H,W], labels: int64}), memory-mapped host side,
  staged to device on a side stream (double-buffered) and normalized/
  augmented on GPU. (JPEG-on-GPU decode is gated on rocJPEG, which is
  not present in this ROCm image.)
"""

from __future__ import annotations

import os
from typing import Any, Iterator, List, Optional, Tuple

import torch

from turboprune_amd.data import augment

IMAGENET_MEAN = torch.tensor([0.485, 0.456, 0.406])
IMAGENET_STD = torch.tensor([0.229, 0.224, 0.225])
IMAGENET_TRAIN_SIZE = 1_281_167
IMAGENET_VAL_SIZE = 50_000


class SyntheticImageNet:
    """GPU-resident synthetic ImageNet-shaped loader (train or val)."""

    def __init__(self, batch_size: int, device: torch.device,
                 train: bool = True, image_size: int = 224,
                 pool_size: int = 1024, seed: int = 0,
                 steps_per_epoch: Optional[int] = None,
                 dtype: torch.dtype = torch.float32,
                 epoch_images: Optional[int] = None):
        self.batch_size = batch_size
        self.device = device
        self.train = train
        self.dtype = dtype
        g = torch.Generator().manual_seed(seed + (0 if train else 1))
        self.pool = torch.randint(0, 256,
                                  (pool_size, 3, image_size, image_size),
                                  dtype=torch.uint8, generator=g).to(device)
        self.labels = torch.randint(0, 1000, (pool_size,),
                                    dtype=torch.int64,
                                    generator=g).to(device)
        if steps_per_epoch is None:
            total = epoch_images or (IMAGENET_TRAIN_SIZE if train
                                     else IMAGENET_VAL_SIZE)
            steps_per_epoch = max(total // batch_size, 1)
        self.steps_per_epoch = steps_per_epoch
        self._mean = IMAGENET_MEAN.to(device)
        self._std = IMAGENET_STD.to(device)
        self._cursor = 0

    def __len__(self) -> int:
        return self.steps_per_epoch

    def __iter__(self) -> Iterator[Tuple[torch.Tensor, torch.Tensor]]:
        n = self.pool.shape[0]
        for _ in range(self.steps_per_epoch):
            idx = torch.arange(self._cursor, self._cursor + self.batch_size,
                               device=self.device) % n
            self._cursor = (self._cursor + self.batch_size) % n
            raw = self.pool[idx]
            flip = None
            if self.train:
                flip = torch.rand(self.batch_size, device=self.device) < 0.5
            x = augment.normalize_u8(raw, self._mean, self._std, self.dtype,
                                     flip)
            yield x, self.labels[idx]


class ShardedImageNet:
    """Pre-decoded uint8 shard reader with async H2D double buffering.

    Shard format: ``{root}/{split}/shard_*.pt`` each a dict with
    ``images`` uint8 [N,3,H,W] and ``labels`` int64 [N]. Sharding across
    ranks: contiguous shard round-robin by rank, per-GPU batch =
    total_batch // world (reference sharding semantics).
    """

    def __init__(self, root: str, split: str, batch_size: int,
                 device: torch.device, train: bool = True,
                 world_size: int = 1, rank: int = 0, seed: int = 0,
                 dtype: torch.dtype = torch.float32,
                 image_size: int = 224):
        self.image_size = image_size
        self.dir = os.path.join(root, split)
        all_paths: List[str] = sorted(
            os.path.join(self.dir, f) for f in os.listdir(self.dir)
            if f.startswith("shard_") and f.endswith(".pt"))
        if not all_paths:
            raise FileNotFoundError(f"no shards under {self.dir}")
        # Every rank must run the SAME number of steps per epoch or the
        # DDP all-reduce deadlocks (ranks with more shards/bigger shards
        # would keep stepping). Each rank reads every shard's header
        # (mmap: metadata only), derives every rank's batch budget, and
        # stops at the global minimum — deterministic, no communication.
        sizes = [self._shard_len(p) for p in all_paths]
        if len(all_paths) < world_size:
            self.paths = [all_paths[rank % len(all_paths)]]
            self.steps_per_epoch = min(sizes) // batch_size
        else:
            self.paths = all_paths[rank::world_size]
            budgets = [sum(s // batch_size
                           for s in sizes[r::world_size])
                       for r in range(world_size)]
            self.steps_per_epoch = min(budgets)
        self.batch_size = batch_size
        self.device = device
        self.train = train
        self.seed = seed
        self.dtype = dtype
        self.epoch = 0
        self._mean = IMAGENET_MEAN.to(device)
        self._std = IMAGENET_STD.to(device)
        self._copy_stream = (torch.cuda.Stream(device)
                             if device.type == "cuda" else None)

    @staticmethod
    def _shard_len(path: str) -> int:
        blob = torch.load(path, map_location="cpu", weights_only=True,
                          mmap=True)
        return int(blob["images"].shape[0])

    def _load_shard(self, path: str):
        blob = torch.load(path, map_location="cpu", weights_only=True,
                          mmap=True)
        return blob["images"], blob["labels"]

    def __iter__(self):
        emitted = 0
        order = list(range(len(self.paths)))
        if self.train:
            g = torch.Generator().manual_seed(self.seed + self.epoch)
            order = torch.randperm(len(order), generator=g).tolist()
        for si in order:
            images, labels = self._load_shard(self.paths[si])
            n = images.shape[0]
            perm = (torch.randperm(n) if self.train else torch.arange(n))
            for i in range(0, n - self.batch_size + 1, self.batch_size):
                idx = perm[i:i + self.batch_size]
                raw_cpu = images[idx].pin_memory() \
                    if self.device.type == "cuda" else images[idx]
                if self._copy_stream is not None:
                    with torch.cuda.stream(self._copy_stream):
                        raw = raw_cpu.to(self.device, non_blocking=True)
                    torch.cuda.current_stream().wait_stream(self._copy_stream)
                else:
                    raw = raw_cpu.to(self.device)
                flip = None
                if self.train:
                    flip = torch.rand(self.batch_size,
                                      device=self.device) < 0.5
                hs, ws = raw.shape[-2], raw.shape[-1]
                if self.train and (hs > self.image_size
                                   or ws > self.image_size):
                    # FFCV-equivalent RandomResizedCrop train pipeline
                    boxes = augment.sample_rrc_boxes(
                        raw.shape[0], hs, ws, device=self.device)
                    x = augment.random_resized_crop(
                        raw, boxes, self._mean, self._std,
                        self.image_size, flip, self.dtype)
                elif not self.train and (hs > self.image_size
                                         or ws > self.image_size):
                    # center crop (val pipeline, ratio-style)
                    t = (hs - self.image_size) // 2
                    l = (ws - self.image_size) // 2
                    raw = raw[..., t:t + self.image_size,
                              l:l + self.image_size].contiguous()
                    x = augment.normalize_u8(raw, self._mean, self._std,
                                             self.dtype, None)
                else:
                    x = augment.normalize_u8(raw, self._mean, self._std,
                                             self.dtype, flip)
                yield x, labels[idx].to(self.device, non_blocking=True)
                emitted += 1
                if emitted >= self.steps_per_epoch:
                    self.epoch += 1
                    return
        self.epoch += 1

    def __len__(self) -> int:
        return self.steps_per_epoch


class ImageNetLoaders:
    """train/test pair honoring cfg.dataset_params
    (dataloader_type: synthetic | native | ffcv-alias)."""

    def __init__(self, cfg: Any, device: torch.device,
                 world_size: int = 1, rank: int = 0,
                 steps_per_epoch: Optional[int] = None):
        total_bs = int(cfg.dataset_params.total_batch_size)
        per_gpu = total_bs // max(world_size, 1)
        seed = int(cfg.select("experiment_params.seed", 0))
        dtype = torch.float32
        kind = cfg.dataset_params.dataloader_type
        root = cfg.dataset_params.data_root_dir
        has_shards = os.path.isdir(os.path.join(str(root), "train"))
        if kind in ("native", "ffcv", "webdataset") and has_shards:
            self.train_loader = ShardedImageNet(
                root, "train", per_gpu, device, True, world_size, rank,
                seed, dtype)
            self.test_loader = ShardedImageNet(
                root, "val", per_gpu, device, False, world_size, rank,
                seed, dtype)
        else:
            # synthetic fallback (no dataset on disk / dataloader_type
            # 'synthetic'): same shapes, same GPU pipeline
            spe = steps_per_epoch
            self.train_loader = SyntheticImageNet(
                per_gpu, device, True, seed=seed + rank,
                steps_per_epoch=spe,
                epoch_images=IMAGENET_TRAIN_SIZE // max(world_size, 1))
            self.test_loader = SyntheticImageNet(
                per_gpu, device, False, seed=seed + rank,
                steps_per_epoch=min(spe or 50, 50),
                epoch_images=IMAGENET_VAL_SIZE // max(world_size, 1))
