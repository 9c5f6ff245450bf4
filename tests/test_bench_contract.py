"""Pin bench.py's driver contract on CPU: the JSON schema the driver
parses, the whole-job (not per-GPU) value semantics, and defaults that
finish within minutes."""

import json
import sys


def _bench():
    sys.path.insert(0, ".")
    import bench
    return bench


def test_result_schema_and_value_semantics():
    bench = _bench()
    import argparse
    ns = argparse.Namespace(gpus=4, steps=30, warmup=10, global_batch=512,
                            model="resnet50", sparsity=0.0, dtype="bf16",
                            no_ddp=False, graph=False)
    r = bench.build_result(ns, world=4, elapsed=3.0, loss_val=6.9,
                           bf16=True)
    required = {"metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling",
                "vs_baseline", "dtype", "data", "config"}
    assert required <= set(r)
    assert r["metric"] == "imagenet_images_per_sec"
    # whole-job aggregate: images/sec counts the GLOBAL batch
    assert r["value"] == round(30 * 512 / 3.0, 1)
    assert r["ms_per_step"] == 100.0
    assert r["scaling"] == "strong"        # global batch fixed as N grows
    assert r["higher_is_better"] is True
    assert r["dtype"] == "bf16"
    assert r["data"] == "synthetic"
    assert r["config"]["global_batch"] == 512
    assert r["config"]["parallelism"] == "dp4"
    json.dumps(r)  # one-line serializable


def test_default_flags_are_short():
    bench = _bench()
    old = sys.argv
    try:
        sys.argv = ["bench.py"]
        args = bench.parse_args()
    finally:
        sys.argv = old
    assert args.gpus == 1
    # 30 timed + 10 warmup steps at the measured ~63 ms/step ≈ 2.5 s of
    # compute — well within the "finishes within minutes" contract
    assert args.steps + args.warmup <= 100
    assert args.global_batch == 512 and args.model == "resnet50"
    assert args.dtype == "bf16"
