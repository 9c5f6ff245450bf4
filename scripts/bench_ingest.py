#!/usr/bin/env python3
"""Measure ShardedImageNet ingest throughput (the
loader must sustain more img/s than the ~8k img/s training rate, in a
measured log, not by assertion).

Builds synthetic 256² shards (unless --root points at real ones), then
times the full GPU pipeline — mmap gather -> pinned ring -> async H2D
-> RandomResizedCrop/flip/normalize HIP kernels — twice:

  bare:    tight loop over the loader (pure ingest ceiling)
  overlap: with a ~40 ms fake GPU compute per batch (does the prefetch
           thread actually hide the gather+copy under compute?)
"""
import argparse
import os
import subprocess
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--root", default="/tmp/ingest_shards")
    p.add_argument("--images", type=int, default=16384)
    p.add_argument("--store-size", type=int, default=256)
    p.add_argument("--batch", type=int, default=512)
    p.add_argument("--compute-ms", type=float, default=40.0)
    args = p.parse_args()

    if not os.path.isdir(os.path.join(args.root, "train")):
        subprocess.run([sys.executable, "scripts/make_shards.py",
                        "--out", args.root, "--split", "train",
                        "--synthetic", str(args.images),
                        "--image-size", str(args.store_size),
                        "--shard-size", "4096"], check=True)

    from turboprune_amd.data.imagenet import ShardedImageNet
    dev = torch.device("cuda:0")
    loader = ShardedImageNet(args.root, "train", args.batch, dev,
                             train=True, dtype=torch.bfloat16)
    print({"shards": len(loader.paths),
           "steps_per_epoch": loader.steps_per_epoch})

    def run(compute_ms):
        # warmup epoch fragment
        n = 0
        for x, y in loader:
            n += x.shape[0]
            if n >= 4 * args.batch:
                break
        torch.cuda.synchronize()
        # dummy compute: batched GEMM sized to ~compute_ms
        a = torch.randn(8192, 8192, device=dev, dtype=torch.bfloat16)
        t0 = time.perf_counter()
        n = 0
        gt = 0.0
        for x, y in loader:
            if compute_ms > 0:
                t1 = time.perf_counter()
                while (time.perf_counter() - t1) * 1e3 < compute_ms:
                    a @ a  # keep the GPU busy like a train step would
                gt += time.perf_counter() - t1
            n += x.shape[0]
        torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        return n, dt, gt

    n, dt, _ = run(0.0)
    print({"mode": "bare", "images": n, "sec": round(dt, 3),
           "img_per_sec": round(n / dt, 1)})
    n, dt, gt = run(args.compute_ms)
    eff = n / (dt - 0)  # loader+compute wall
    print({"mode": f"overlap_{args.compute_ms}ms", "images": n,
           "sec": round(dt, 3), "compute_sec": round(gt, 3),
           "img_per_sec_wall": round(eff, 1),
           "loader_overhead_sec": round(dt - gt, 3)})


if __name__ == "__main__":
    main()
