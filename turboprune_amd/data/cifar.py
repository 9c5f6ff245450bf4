"""Airbench-style GPU-resident CIFAR loader.

Design follows the reference's CifarLoader (utils/dataset.py:101-256):
the whole dataset lives on the GPU as a uint8 tensor; the first iteration
normalizes and reflect-pads once; each epoch applies GPU augmentation
(random-translate crop, alternating-epoch flip, optional cutout) and
yields device-resident batches via randperm slicing — zero host<->device
traffic per epoch.

Data source: a cached tensor file ``{dataset}_train.pt`` /
``{dataset}_test.pt`` under ``data_root_dir`` (dict with uint8
``images`` [N,3,32,32] and int64 ``labels``), written under a FileLock
(reference: dataset.py:127-149). There is no network in the target
environment; when the cache is absent a deterministic synthetic dataset
of the same shape is generated (and the loader records
``synthetic=True``).
"""

from __future__ import annotations

import os
from typing import Any, Iterator, Optional, Tuple

import torch
from filelock import FileLock

from turboprune_amd.data import augment

CIFAR_MEAN = torch.tensor([0.4914, 0.4822, 0.4465])
CIFAR_STD = torch.tensor([0.2470, 0.2435, 0.2616])


def _cache_path(root: str, dataset: str, train: bool) -> str:
    split = "train" if train else "test"
    return os.path.join(root, f"{dataset.lower()}_{split}.pt")


def _load_or_synthesize(root: str, dataset: str, train: bool,
                        synthetic_size: int, seed: int,
                        allow_synthetic: bool = False):
    path = _cache_path(root, dataset, train)
    if os.path.exists(path):
        with FileLock(path + ".lock"):
            blob = torch.load(path, map_location="cpu", weights_only=True)
        return blob["images"], blob["labels"], False
    if not allow_synthetic:
        # never silently train on noise — the
        # reference pipeline fails fast on missing data too.
        raise FileNotFoundError(
            f"no {dataset} cache at {path}. Provide the tensor cache, or "
            f"opt into synthetic data (+dataset_params.synthetic_size=N "
            f"or TURBOPRUNE_SYNTHETIC_CIFAR=1).")
    n_classes = 100 if dataset.upper() == "CIFAR100" else 10
    g = torch.Generator().manual_seed(seed + (0 if train else 1))
    images = torch.randint(0, 256, (synthetic_size, 3, 32, 32),
                           dtype=torch.uint8, generator=g)
    labels = torch.randint(0, n_classes, (synthetic_size,),
                           dtype=torch.int64, generator=g)
    return images, labels, True


class CifarLoader:
    def __init__(self, root: str, dataset: str = "CIFAR10",
                 train: bool = True, batch_size: int = 512,
                 aug: Optional[dict] = None, altflip: bool = True,
                 device: Optional[torch.device] = None,
                 synthetic_size: int = 2048, seed: int = 0,
                 drop_last: Optional[bool] = None,
                 allow_synthetic: Optional[bool] = None):
        self.device = device or torch.device(
            "cuda" if torch.cuda.is_available() else "cpu")
        if allow_synthetic is None:
            allow_synthetic = (
                os.environ.get("TURBOPRUNE_SYNTHETIC_CIFAR") == "1")
        images_u8, labels, self.synthetic = _load_or_synthesize(
            root, dataset, train, synthetic_size, seed, allow_synthetic)
        if self.synthetic and train:
            print(f"[turboprune] WARNING: no {dataset} cache under "
                  f"{root}; using SYNTHETIC random data "
                  f"(explicitly allowed).", flush=True)
        self.images_u8 = images_u8.to(self.device)
        self.labels = labels.to(self.device)
        self.batch_size = batch_size
        self.train = train
        self.aug = dict(aug or {})
        self.altflip = altflip
        self.drop_last = train if drop_last is None else drop_last
        self.epoch = 0
        self._normalized: Optional[torch.Tensor] = None
        self._padded: Optional[torch.Tensor] = None

    def __len__(self) -> int:
        n = self.images_u8.shape[0]
        bs = self.batch_size
        return n // bs if self.drop_last else (n + bs - 1) // bs

    @property
    def num_samples(self) -> int:
        return self.images_u8.shape[0]

    def _prepare(self) -> None:
        """One-time whole-set normalization (+ reflect pad for translate)
        (reference: dataset.py:192-201)."""
        if self._normalized is not None:
            return
        x = augment.normalize_u8(self.images_u8, CIFAR_MEAN, CIFAR_STD,
                                 torch.float32)
        self._normalized = x
        pad = int(self.aug.get("translate", 0))
        if self.train and pad > 0:
            self._padded = torch.nn.functional.pad(
                x, (pad, pad, pad, pad), mode="reflect")

    def __iter__(self) -> Iterator[Tuple[torch.Tensor, torch.Tensor]]:
        self._prepare()
        n = self.num_samples
        if not self.train:
            for i in range(0, n, self.batch_size):
                yield (self._normalized[i:i + self.batch_size],
                       self.labels[i:i + self.batch_size])
            return

        pad = int(self.aug.get("translate", 0))
        if pad > 0:
            shifts = torch.randint(0, 2 * pad + 1, (n, 2), device=self.device)
            images = augment.batch_crop_translate(self._padded, 32, shifts)
        else:
            images = self._normalized.clone()

        if self.aug.get("flip", False):
            if self.altflip:
                # alternating-epoch deterministic flip of each half
                # (reference: dataset.py:209-215)
                half = torch.arange(n, device=self.device) % 2 == self.epoch % 2
                images = augment.batch_flip_lr(images, half)
            else:
                flip = torch.rand(n, device=self.device) < 0.5
                images = augment.batch_flip_lr(images, flip)

        cutout = int(self.aug.get("cutout", 0))
        if cutout > 0:
            augment.batch_cutout(images, cutout)

        perm = torch.randperm(n, device=self.device)
        limit = (n // self.batch_size) * self.batch_size if self.drop_last else n
        for i in range(0, limit, self.batch_size):
            idx = perm[i:i + self.batch_size]
            yield images[idx], self.labels[idx]
        self.epoch += 1


class AirbenchLoaders:
    """train/test pair with the reference's default augmentation
    (flip + translate 2, altflip; reference: dataset.py:229-256)."""

    def __init__(self, cfg: Any, device: Optional[torch.device] = None,
                 synthetic_size: Optional[int] = None):
        root = cfg.dataset_params.data_root_dir
        dataset = cfg.dataset_params.dataset_name
        bs = int(cfg.dataset_params.total_batch_size)
        seed = int(cfg.select("experiment_params.seed", 0))
        explicit = (synthetic_size is not None
                    or cfg.select("dataset_params.synthetic_size", None)
                    is not None)
        if synthetic_size is None:
            synthetic_size = int(cfg.select("dataset_params.synthetic_size",
                                            2048))
        # an explicitly configured synthetic_size is the opt-in; without
        # it CifarLoader falls through to the env-var/fail-fast check
        allow = True if explicit else None
        self.train_loader = CifarLoader(
            root, dataset, train=True, batch_size=bs,
            aug={"flip": True, "translate": 2}, altflip=True, device=device,
            synthetic_size=synthetic_size, seed=seed, allow_synthetic=allow)
        self.test_loader = CifarLoader(
            root, dataset, train=False, batch_size=bs, device=device,
            synthetic_size=max(synthetic_size // 4, 256), seed=seed,
            allow_synthetic=allow)
