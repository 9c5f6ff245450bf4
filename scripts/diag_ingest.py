#!/usr/bin/env python3
"""Box-side ingest diagnosis: where do the ~200 ms/batch go?
(Locally the warm mmap gather runs at 17 GB/s; the box measured
~0.45 GB/s.) Times each pipeline stage in isolation."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

ROOT = "/tmp/ingest_diag"


def main():
    import subprocess
    if not os.path.isdir(os.path.join(ROOT, "train")):
        subprocess.run([sys.executable, "scripts/make_shards.py",
                        "--out", ROOT, "--split", "train",
                        "--synthetic", "8192", "--image-size", "256",
                        "--shard-size", "4096"], check=True)
    print({"nproc": os.cpu_count(), "torch_threads": torch.get_num_threads()})
    path = os.path.join(ROOT, "train", "shard_00000.pt")
    fd = os.open(path, os.O_RDONLY)
    os.posix_fadvise(fd, 0, 0, os.POSIX_FADV_WILLNEED)
    os.close(fd)
    blob = torch.load(path, map_location="cpu", weights_only=True,
                      mmap=True)
    imgs = blob["images"]
    nb = imgs.numel()

    t0 = time.perf_counter()
    s = imgs.sum(dtype=torch.int64)  # force page-in
    dt = time.perf_counter() - t0
    print({"page_in_GBs": round(nb / dt / 1e9, 2)})

    buf = torch.empty(512, 3, 256, 256, dtype=torch.uint8)
    pbuf = torch.empty(512, 3, 256, 256, dtype=torch.uint8,
                       pin_memory=torch.cuda.is_available())
    for name, out in (("plain", buf), ("pinned", pbuf)):
        for trial in range(2):
            idx = torch.randperm(4096)[:512]
            t0 = time.perf_counter()
            torch.index_select(imgs, 0, idx, out=out)
            dt = time.perf_counter() - t0
            print({"gather_to": name, "trial": trial,
                   "GBs": round(out.numel() / dt / 1e9, 2),
                   "ms": round(dt * 1e3, 1)})
    if torch.cuda.is_available():
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(5):
            d = pbuf.to("cuda", non_blocking=True)
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / 5
        print({"h2d_GBs": round(pbuf.numel() / dt / 1e9, 2)})
        from turboprune_amd.data import augment
        mean = torch.tensor([0.485, 0.456, 0.406], device="cuda")
        std = torch.tensor([0.229, 0.224, 0.225], device="cuda")
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(5):
            boxes = augment.sample_rrc_boxes(512, 256, 256,
                                             device=torch.device("cuda"))
            x = augment.random_resized_crop(d, boxes, mean, std, 224,
                                            None, torch.bfloat16)
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / 5
        print({"rrc_normalize_ms": round(dt * 1e3, 2)})


if __name__ == "__main__":
    main()
