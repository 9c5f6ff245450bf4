#!/usr/bin/env python3
"""Create turboprune_amd ImageNet shard files (the FFCV-beton-equivalent
record format: .pt shards of pre-decoded uint8 images + labels).

From a directory tree of class-subdirectories of images (requires PIL,
optional in this environment), or synthetically for pipeline testing:

    python scripts/make_shards.py --out /data/imagenet_shards \
        --synthetic 10000 --image-size 224 --shard-size 2048
"""

import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def write_shard(path, images, labels):
    torch.save({"images": images, "labels": labels}, path)


def make_synthetic(out_dir, split, n, image_size, shard_size, seed=0):
    os.makedirs(os.path.join(out_dir, split), exist_ok=True)
    g = torch.Generator().manual_seed(seed)
    idx = 0
    shard = 0
    while idx < n:
        m = min(shard_size, n - idx)
        images = torch.randint(0, 256, (m, 3, image_size, image_size),
                               dtype=torch.uint8, generator=g)
        labels = torch.randint(0, 1000, (m,), dtype=torch.int64,
                               generator=g)
        write_shard(os.path.join(out_dir, split, f"shard_{shard:05d}.pt"),
                    images, labels)
        idx += m
        shard += 1
    print(f"wrote {shard} shards ({n} images) under {out_dir}/{split}")


def make_from_images(out_dir, split, src_dir, store_size, shard_size,
                     workers=0):
    """Decode a class-subdir JPEG tree (PIL/libjpeg on CPU — rocJPEG is
    absent from this ROCm image) into uint8 shards at ``store_size``²
    (shorter-side resize + center crop), leaving RandomResizedCrop
    headroom for the GPU train pipeline (store 256 -> train crops 224,
    the FFCV 'max side' idea; reference: utils/dataset.py:385-400)."""
    try:
        from PIL import Image  # noqa
    except ImportError:
        raise SystemExit("PIL not available in this environment; use "
                         "--synthetic or pre-decode elsewhere")
    import numpy as np

    def decode_one(path):
        img = Image.open(path).convert("RGB")
        w, h = img.size
        s = store_size / min(w, h)
        img = img.resize((max(round(w * s), store_size),
                          max(round(h * s), store_size)), Image.BILINEAR)
        w, h = img.size
        l, t = (w - store_size) // 2, (h - store_size) // 2
        img = img.crop((l, t, l + store_size, t + store_size))
        return torch.from_numpy(np.asarray(img)).permute(2, 0, 1) \
            .contiguous()

    classes = sorted(d for d in os.listdir(src_dir)
                     if os.path.isdir(os.path.join(src_dir, d)))
    os.makedirs(os.path.join(out_dir, split), exist_ok=True)
    jobs = [(os.path.join(src_dir, cls, f), ci)
            for ci, cls in enumerate(classes)
            for f in sorted(os.listdir(os.path.join(src_dir, cls)))]
    buf_imgs, buf_lbls, shard, total = [], [], 0, 0

    def flush():
        nonlocal buf_imgs, buf_lbls, shard
        if buf_imgs:
            write_shard(os.path.join(out_dir, split,
                                     f"shard_{shard:05d}.pt"),
                        torch.stack(buf_imgs), torch.tensor(buf_lbls))
            buf_imgs, buf_lbls = [], []
            shard += 1

    if workers > 1:
        from concurrent.futures import ThreadPoolExecutor  # PIL drops GIL
        with ThreadPoolExecutor(workers) as pool:
            for arr, ci in zip(pool.map(lambda j: decode_one(j[0]), jobs),
                               (j[1] for j in jobs)):
                buf_imgs.append(arr)
                buf_lbls.append(ci)
                total += 1
                if len(buf_imgs) == shard_size:
                    flush()
    else:
        for path, ci in jobs:
            buf_imgs.append(decode_one(path))
            buf_lbls.append(ci)
            total += 1
            if len(buf_imgs) == shard_size:
                flush()
    flush()
    print(f"wrote {shard} shards ({total} images, {store_size}²) "
          f"under {out_dir}/{split}")


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--out", required=True)
    p.add_argument("--split", default="train")
    p.add_argument("--src", default=None,
                   help="class-subdir image tree (requires PIL)")
    p.add_argument("--synthetic", type=int, default=0,
                   help="generate N synthetic images instead")
    p.add_argument("--image-size", type=int, default=224,
                   help="synthetic image size")
    p.add_argument("--store-size", type=int, default=256,
                   help="--src stored crop size (>224 leaves RRC headroom)")
    p.add_argument("--shard-size", type=int, default=2048)
    p.add_argument("--workers", type=int, default=os.cpu_count() or 1)
    args = p.parse_args()
    if args.synthetic:
        make_synthetic(args.out, args.split, args.synthetic,
                       args.image_size, args.shard_size)
    elif args.src:
        make_from_images(args.out, args.split, args.src, args.store_size,
                         args.shard_size, args.workers)
    else:
        raise SystemExit("need --src or --synthetic")


if __name__ == "__main__":
    main()
