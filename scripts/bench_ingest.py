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

    # full warm pass: page the shards into the OS cache so the timed
    # passes measure the pipeline, not first-touch disk reads (FFCV's
    # os_cache behaves the same way)
    t0 = time.perf_counter()
    n = sum(x.shape[0] for x, _ in loader)
    torch.cuda.synchronize()
    print({"mode": "cold_first_epoch", "images": n,
           "sec": round(time.perf_counter() - t0, 3),
           "img_per_sec": round(n / (time.perf_counter() - t0), 1)})

    # fake train step: a fixed GEMM loop ~compute_ms of GPU work per
    # batch, enqueued once per batch (bounded queue depth — the point is
    # to see whether the prefetch thread hides gather+H2D under it)
    a = torch.randn(6144, 6144, device=dev, dtype=torch.bfloat16)
    torch.cuda.synchronize()
    t1 = time.perf_counter()
    for _ in range(8):
        a = a @ a * 1e-3
    torch.cuda.synchronize()
    per_gemm_ms = (time.perf_counter() - t1) / 8 * 1e3
    gemms = max(int(args.compute_ms / per_gemm_ms), 1)

    o = torch.empty_like(a)

    def run(compute):
        t0 = time.perf_counter()
        n = 0
        for x, y in loader:
            if compute:
                for _ in range(gemms):
                    torch.mm(a, a, out=o)
            n += x.shape[0]
        torch.cuda.synchronize()
        return n, time.perf_counter() - t0

    n, dt = run(False)
    print({"mode": "warm_bare", "images": n, "sec": round(dt, 3),
           "img_per_sec": round(n / dt, 1)})
    n, dt = run(True)
    compute_total = gemms * per_gemm_ms * (n // args.batch) / 1e3
    print({"mode": f"warm_overlap~{args.compute_ms}ms", "images": n,
           "sec": round(dt, 3),
           "compute_sec_est": round(compute_total, 3),
           "img_per_sec_wall": round(n / dt, 1),
           "loader_overhead_sec": round(dt - compute_total, 3)})


if __name__ == "__main__":
    main()
