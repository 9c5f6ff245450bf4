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


def make_from_images(out_dir, split, src_dir, image_size, shard_size):
    try:
        from PIL import Image  # noqa
    except ImportError:
        raise SystemExit("PIL not available in this environment; use "
                         "--synthetic or pre-decode elsewhere")
    import numpy as np
    classes = sorted(d for d in os.listdir(src_dir)
                     if os.path.isdir(os.path.join(src_dir, d)))
    os.makedirs(os.path.join(out_dir, split), exist_ok=True)
    buf_imgs, buf_lbls, shard = [], [], 0
    for ci, cls in enumerate(classes):
        cdir = os.path.join(src_dir, cls)
        for fname in sorted(os.listdir(cdir)):
            img = Image.open(os.path.join(cdir, fname)).convert("RGB")
            img = img.resize((image_size, image_size), Image.BILINEAR)
            arr = torch.from_numpy(np.asarray(img)).permute(2, 0, 1)
            buf_imgs.append(arr.contiguous())
            buf_lbls.append(ci)
            if len(buf_imgs) == shard_size:
                write_shard(
                    os.path.join(out_dir, split, f"shard_{shard:05d}.pt"),
                    torch.stack(buf_imgs), torch.tensor(buf_lbls))
                buf_imgs, buf_lbls = [], []
                shard += 1
    if buf_imgs:
        write_shard(os.path.join(out_dir, split, f"shard_{shard:05d}.pt"),
                    torch.stack(buf_imgs), torch.tensor(buf_lbls))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--out", required=True)
    p.add_argument("--split", default="train")
    p.add_argument("--src", default=None,
                   help="class-subdir image tree (requires PIL)")
    p.add_argument("--synthetic", type=int, default=0,
                   help="generate N synthetic images instead")
    p.add_argument("--image-size", type=int, default=224)
    p.add_argument("--shard-size", type=int, default=2048)
    args = p.parse_args()
    if args.synthetic:
        make_synthetic(args.out, args.split, args.synthetic,
                       args.image_size, args.shard_size)
    elif args.src:
        make_from_images(args.out, args.split, args.src, args.image_size,
                         args.shard_size)
    else:
        raise SystemExit("need --src or --synthetic")


if __name__ == "__main__":
    main()
