import os

import torch

from turboprune_amd.config import compose
from turboprune_amd.data import (AirbenchLoaders, CifarLoader,
                                 SyntheticImageNet)
from turboprune_amd.data import augment
from turboprune_amd.data.imagenet import IMAGENET_MEAN, IMAGENET_STD


def _cfg(tmp_path):
    return compose("cifar10_er_erk", [
        f"dataset_params.data_root_dir={tmp_path}/data",
        "dataset_params.total_batch_size=64",
        "+dataset_params.synthetic_size=256",
    ])


def test_cifar_loader_shapes(tmp_path):
    cfg = _cfg(tmp_path)
    pair = AirbenchLoaders(cfg, device=torch.device("cpu"))
    batches = list(pair.train_loader)
    assert len(batches) == 4  # 256/64, drop_last
    x, y = batches[0]
    assert x.shape == (64, 3, 32, 32)
    assert x.dtype == torch.float32
    assert y.shape == (64,)
    assert pair.train_loader.synthetic  # no real CIFAR on disk


def test_cifar_loader_epoch_variation(tmp_path):
    cfg = _cfg(tmp_path)
    loader = AirbenchLoaders(cfg, device=torch.device("cpu")).train_loader
    torch.manual_seed(0)
    e0 = torch.cat([x.sum(dim=(1, 2, 3)) for x, _ in loader])
    e1 = torch.cat([x.sum(dim=(1, 2, 3)) for x, _ in loader])
    # augmentation + reshuffle change the epoch content
    assert not torch.allclose(e0.sort().values, e1.sort().values)


def test_cifar_test_loader_deterministic(tmp_path):
    cfg = _cfg(tmp_path)
    loader = AirbenchLoaders(cfg, device=torch.device("cpu")).test_loader
    a = torch.cat([x for x, _ in loader])
    b = torch.cat([x for x, _ in loader])
    assert torch.equal(a, b)


def test_normalize_u8_oracle():
    imgs = torch.randint(0, 256, (4, 3, 8, 8), dtype=torch.uint8)
    mean = IMAGENET_MEAN
    std = IMAGENET_STD
    out = augment.normalize_u8(imgs, mean, std)
    ref = (imgs.float() / 255.0 - mean.view(1, 3, 1, 1)) / std.view(1, 3, 1, 1)
    assert torch.allclose(out, ref, atol=1e-6)


def test_normalize_u8_flip():
    imgs = torch.randint(0, 256, (2, 3, 4, 4), dtype=torch.uint8)
    flip = torch.tensor([True, False])
    out = augment.normalize_u8(imgs, torch.zeros(3), torch.ones(3),
                               flip=flip)
    assert torch.allclose(out[0], torch.flip(imgs[0].float() / 255, [-1]))
    assert torch.allclose(out[1], imgs[1].float() / 255)


def test_batch_crop_translate():
    x = torch.arange(2 * 1 * 6 * 6, dtype=torch.float32).reshape(2, 1, 6, 6)
    shifts = torch.tensor([[0, 0], [2, 2]])
    out = augment.batch_crop_translate(x, 4, shifts)
    assert out.shape == (2, 1, 4, 4)
    assert torch.equal(out[0, 0], x[0, 0, 0:4, 0:4])
    assert torch.equal(out[1, 0], x[1, 0, 2:6, 2:6])


def test_batch_cutout():
    x = torch.ones(8, 3, 16, 16)
    augment.batch_cutout(x, 4)
    # every image lost some pixels, none lost everything
    per_image = x.sum(dim=(1, 2, 3))
    assert torch.all(per_image < 3 * 256)
    assert torch.all(per_image > 0)


def test_synthetic_imagenet_sharding_math():
    dev = torch.device("cpu")
    loader = SyntheticImageNet(batch_size=64, device=dev, train=True,
                               image_size=32, pool_size=128,
                               steps_per_epoch=3)
    batches = list(loader)
    assert len(batches) == 3
    x, y = batches[0]
    assert x.shape == (64, 3, 32, 32)
    assert y.min() >= 0 and y.max() < 1000


def test_imagenet_loaders_synthetic_fallback(tmp_path):
    cfg = compose("bench_resnet50_imagenet", [
        "dataset_params.total_batch_size=32",
        "+dataset_params.steps_per_epoch=2",
        f"dataset_params.data_root_dir={tmp_path}",
    ])
    from turboprune_amd.data import ImageNetLoaders
    pair = ImageNetLoaders(cfg, torch.device("cpu"), world_size=4, rank=1,
                           steps_per_epoch=2)
    x, y = next(iter(pair.train_loader))
    assert x.shape == (8, 3, 224, 224)  # 32 total / 4 ranks


def test_sharded_imagenet_reader(tmp_path):
    import subprocess
    import sys
    subprocess.run([sys.executable, "scripts/make_shards.py",
                    "--out", str(tmp_path), "--split", "train",
                    "--synthetic", "200", "--image-size", "32",
                    "--shard-size", "64"], check=True,
                   cwd=os.path.dirname(os.path.dirname(
                       os.path.abspath(__file__))))
    from turboprune_amd.data import ShardedImageNet
    loader = ShardedImageNet(str(tmp_path), "train", batch_size=16,
                             device=torch.device("cpu"), train=True,
                             world_size=1, rank=0)
    batches = list(loader)
    assert len(batches) >= 10  # 200 imgs, batch 16, drop-last per shard
    x, y = batches[0]
    assert x.shape == (16, 3, 32, 32) and x.dtype == torch.float32
    assert y.dtype == torch.int64

    # rank sharding: 2 ranks see disjoint shard subsets
    l0 = ShardedImageNet(str(tmp_path), "train", 16, torch.device("cpu"),
                         True, world_size=2, rank=0)
    l1 = ShardedImageNet(str(tmp_path), "train", 16, torch.device("cpu"),
                         True, world_size=2, rank=1)
    assert set(l0.paths).isdisjoint(set(l1.paths))


def test_sample_rrc_boxes():
    torch.manual_seed(0)
    boxes = augment.sample_rrc_boxes(64, 256, 256)
    assert boxes.shape == (64, 4)
    t, l, h, w = boxes.unbind(1)
    assert (t >= 0).all() and (l >= 0).all()
    assert ((t + h) <= 256).all() and ((l + w) <= 256).all()
    assert (h > 0).all() and (w > 0).all()
    # aspect ratio within [3/4, 4/3] (allow rounding slack)
    ar = w.float() / h.float()
    assert (ar > 0.7).all() and (ar < 1.43).all()


def test_random_resized_crop_cpu_oracle():
    torch.manual_seed(1)
    imgs = torch.randint(0, 256, (3, 3, 64, 64), dtype=torch.uint8)
    boxes = torch.tensor([[0, 0, 64, 64], [10, 10, 32, 32], [0, 0, 48, 64]])
    mean = torch.tensor([0.485, 0.456, 0.406])
    std = torch.tensor([0.229, 0.224, 0.225])
    out = augment.random_resized_crop(imgs, boxes, mean, std, out_size=32)
    assert out.shape == (3, 3, 32, 32)
    # full-image box == plain resize
    ref = torch.nn.functional.interpolate(
        imgs[0:1].float(), size=(32, 32), mode="bilinear",
        align_corners=False) / 255.0
    ref = (ref - mean.view(1, 3, 1, 1)) / std.view(1, 3, 1, 1)
    assert torch.allclose(out[0], ref[0], atol=1e-5)


def test_sharded_reader_rrc_path(tmp_path):
    """Shards stored at 64px, loader at 32px: train uses RRC, val center
    crop."""
    import subprocess
    import sys
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    for split in ("train", "val"):
        subprocess.run([sys.executable, "scripts/make_shards.py",
                        "--out", str(tmp_path), "--split", split,
                        "--synthetic", "64", "--image-size", "64",
                        "--shard-size", "64"], check=True, cwd=root)
    from turboprune_amd.data import ShardedImageNet
    tr = ShardedImageNet(str(tmp_path), "train", 16, torch.device("cpu"),
                         True, image_size=32)
    x, y = next(iter(tr))
    assert x.shape == (16, 3, 32, 32)
    va = ShardedImageNet(str(tmp_path), "val", 16, torch.device("cpu"),
                         False, image_size=32)
    xv, yv = next(iter(va))
    assert xv.shape == (16, 3, 32, 32)


def test_sharded_equal_steps_across_ranks(tmp_path):
    """Unequal shard counts/sizes must NOT give ranks different step
    counts (per-step DDP all-reduce would deadlock): every rank stops at
    the global minimum batch budget, derived deterministically from all
    shard headers."""
    import torch

    from turboprune_amd.data.imagenet import ShardedImageNet
    root = tmp_path / "inet"
    (root / "train").mkdir(parents=True)
    # 3 shards of sizes 10, 10, 6 -> world 2: rank0 gets [10, 6]=4
    # batches@bs4, rank1 gets [10]=2 batches -> both must run 2
    for i, n in enumerate([10, 10, 6]):
        torch.save({"images": torch.randint(0, 255, (n, 3, 8, 8),
                                            dtype=torch.uint8),
                    "labels": torch.randint(0, 10, (n,))},
                   root / "train" / f"shard_{i:03d}.pt")
    counts = []
    for rank in range(2):
        ld = ShardedImageNet(str(root), "train", batch_size=4,
                             device=torch.device("cpu"), train=True,
                             world_size=2, rank=rank, image_size=8)
        counts.append(sum(1 for _ in ld))
        assert len(ld) == counts[-1]
    assert counts[0] == counts[1] == 2

    # more ranks than shards: everyone gets one shard, equal budget
    for rank in range(4):
        ld = ShardedImageNet(str(root), "train", batch_size=4,
                             device=torch.device("cpu"), train=True,
                             world_size=4, rank=rank, image_size=8)
        assert sum(1 for _ in ld) == 1  # min shard 6 // 4 = 1


def test_make_shards_from_real_jpegs(tmp_path):
    """Real-JPEG ingest path (VERDICT r01 missing #2): PIL-decoded JPEG
    tree -> uint8 shards at store-size with shorter-side resize + center
    crop -> ShardedImageNet reads them back."""
    import subprocess
    import sys

    from PIL import Image
    src = tmp_path / "jpegs"
    rng = torch.Generator().manual_seed(3)
    for ci, cls in enumerate(["n01440764", "n01443537"]):
        d = src / cls
        d.mkdir(parents=True)
        for i in range(6):
            h = int(torch.randint(40, 90, (1,), generator=rng))
            w = int(torch.randint(40, 90, (1,), generator=rng))
            arr = torch.randint(0, 256, (h, w, 3), dtype=torch.uint8,
                                generator=rng).numpy()
            Image.fromarray(arr).save(d / f"img_{i}.jpg", quality=90)
    out = tmp_path / "shards"
    subprocess.run([sys.executable, "scripts/make_shards.py",
                    "--out", str(out), "--split", "train",
                    "--src", str(src), "--store-size", "32",
                    "--shard-size", "8", "--workers", "2"], check=True,
                   cwd=os.path.dirname(os.path.dirname(
                       os.path.abspath(__file__))))
    from turboprune_amd.data import ShardedImageNet
    loader = ShardedImageNet(str(out), "train", batch_size=4,
                             device=torch.device("cpu"), train=True)
    x, y = next(iter(loader))
    assert x.shape == (4, 3, 32, 32)
    assert y.min() >= 0 and y.max() <= 1


def test_sharded_split_when_fewer_shards_than_ranks(tmp_path):
    """with 1 shard and 2 ranks, the shard is split by index
    range — ranks must not read identical data."""
    import subprocess
    import sys
    subprocess.run([sys.executable, "scripts/make_shards.py",
                    "--out", str(tmp_path), "--split", "train",
                    "--synthetic", "64", "--image-size", "16",
                    "--shard-size", "64"], check=True,
                   cwd=os.path.dirname(os.path.dirname(
                       os.path.abspath(__file__))))
    from turboprune_amd.data import ShardedImageNet
    l0 = ShardedImageNet(str(tmp_path), "train", 8, torch.device("cpu"),
                         train=False, world_size=2, rank=0)
    l1 = ShardedImageNet(str(tmp_path), "train", 8, torch.device("cpu"),
                         train=False, world_size=2, rank=1)
    assert l0.steps_per_epoch == l1.steps_per_epoch == 4  # 32 imgs each
    y0 = torch.cat([y for _, y in l0])
    y1 = torch.cat([y for _, y in l1])
    # index ranges are disjoint halves of the same shard
    blob = torch.load(os.path.join(str(tmp_path), "train",
                                   "shard_00000.pt"), weights_only=True)
    assert torch.equal(y0, blob["labels"][:32])
    assert torch.equal(y1, blob["labels"][32:])


def test_imagenet_loaders_fail_fast_without_shards(tmp_path):
    """a real dataloader_type with no shards must
    raise, not silently train on noise."""
    import pytest
    cfg = compose("bench_resnet50_imagenet", [
        "dataset_params.dataloader_type=native",
        f"dataset_params.data_root_dir={tmp_path}/nope",
    ])
    from turboprune_amd.data import ImageNetLoaders
    with pytest.raises(FileNotFoundError):
        ImageNetLoaders(cfg, torch.device("cpu"))
    # explicit opt-in downgrades to synthetic with a loud warning
    cfg2 = compose("bench_resnet50_imagenet", [
        "dataset_params.dataloader_type=native",
        f"dataset_params.data_root_dir={tmp_path}/nope",
        "+dataset_params.allow_synthetic_fallback=true",
        "+dataset_params.steps_per_epoch=1",
    ])
    pair = ImageNetLoaders(cfg2, torch.device("cpu"), steps_per_epoch=1)
    assert pair.synthetic


def test_cifar_loader_fail_fast(tmp_path, monkeypatch):
    """CifarLoader without cache and without explicit synthetic opt-in
    must raise."""
    import pytest
    monkeypatch.delenv("TURBOPRUNE_SYNTHETIC_CIFAR", raising=False)
    with pytest.raises(FileNotFoundError):
        CifarLoader(str(tmp_path / "none"), "CIFAR10", train=True,
                    batch_size=8)
