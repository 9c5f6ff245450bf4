"""Compare a bench.py JSON line against the recorded round-1 numbers.

    python bench.py ... | tee /tmp/b.json
    python scripts/check_perf_regression.py /tmp/b.json

Exit 1 if the measured throughput regresses more than --tol (default 3%)
below the stored floor for that model/global-batch. Floors live in this
file on purpose (updated by hand when a promotion lands, so an
accidental regression cannot silently ratchet them down)."""

import argparse
import json
import sys

# (model, global_batch, n_gpus) -> images/sec floor (round-2 measured:
# auto conv dispatch 8151-8175, DeiT with per-shape GEMM routing 7239)
FLOORS = {
    ("resnet50", 512, 1): 8300.0,
    ("deit_small", 256, 1): 7420.0,
}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("path")
    ap.add_argument("--tol", type=float, default=0.03)
    args = ap.parse_args()
    line = None
    with open(args.path) as f:
        for ln in f:
            ln = ln.strip()
            if ln.startswith("{") and '"metric"' in ln:
                line = ln
    if line is None:
        print("no bench JSON line found")
        sys.exit(2)
    r = json.loads(line)
    key = (r["config"]["model"], r["config"]["global_batch"], r["n_gpus"])
    floor = FLOORS.get(key)
    if floor is None:
        print(f"no floor recorded for {key}; measured {r['value']}")
        sys.exit(0)
    ok = r["value"] >= floor * (1.0 - args.tol)
    print(json.dumps({"key": list(key), "measured": r["value"],
                      "floor": floor, "ok": ok}))
    sys.exit(0 if ok else 1)


if __name__ == "__main__":
    main()
