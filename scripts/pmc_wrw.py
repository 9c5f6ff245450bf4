#!/usr/bin/env python3
"""Kernel-only wrw workload for PMC counter collection (guide: measure,
don't guess). Run COUNTERS-ONLY per the pool rule:

  cd /tmp && export TMPDIR=/tmp
  rocprofv3 --pmc SQ_VALU_MFMA_BUSY_CYCLES SQ_INSTS_LDS SQ_LDS_BANK_CONFLICT \
      SQ_INSTS_VALU -d gpurun_out/pmc_wrw -- python scripts/pmc_wrw.py

Two contrasting shapes: small-C deep-M (layer1 3x3) and big-C shallow-M
(layer4 3x3). 20 launches each so the dispatch table averages cleanly.
The env knob TURBOPRUNE_WRW selects the kernel generation (default v2).
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from turboprune_amd.ops._backend import extension  # noqa: E402


def main():
    ext = extension()
    dev = "cuda:0"
    for (cin, cout, k, s, hi) in [(64, 64, 3, 1, 56), (512, 512, 3, 1, 7)]:
        p = k // 2
        torch.manual_seed(cin)
        x = (torch.rand(512, cin, hi, hi, device=dev) - .5).bfloat16() \
            .to(memory_format=torch.channels_last)
        ho = (hi + 2 * p - k) // s + 1
        gy = torch.randn(512, cout, ho, ho, device=dev).bfloat16() \
            .to(memory_format=torch.channels_last)
        for _ in range(3):
            ext.conv2d_implicit_wrw(gy, x, k, k, s, p)
        torch.cuda.synchronize()
        for _ in range(20):
            ext.conv2d_implicit_wrw(gy, x, k, k, s, p)
        torch.cuda.synchronize()
        print("done", (cin, cout, k, s, hi))


if __name__ == "__main__":
    main()
