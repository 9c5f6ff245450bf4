#!/bin/bash
# Round-2 opening gpurun call: validate everything drafted off-device in
# round 1, in one box session. Each step is independently timeout-capped
# so one bad kernel cannot eat the call. Run as:
#   /usr/local/graft/bin/gpurun --timeout 2400 -- 'bash scripts/round2_device_checks.sh'
# (realistic runtime ~20-30 min: the per-step timeouts are caps, not
#  expected durations; most cost is per-invocation torch/model init)
# then read gpurun_out/r2_checks/*.log and promote what passed:
#   gemm256 PASS  -> wire TURBOPRUNE_GEMM256 into masked_linear dispatch
#   attn PASS     -> flip TURBOPRUNE_ATTN=native for DeiT bench, A/B
#   conv PASS     -> TURBOPRUNE_CONV=native bench A/B (the headline lever)
#   multi-sgd     -> TURBOPRUNE_MULTI_SGD=1 bench A/B at small batch
set -u
OUT=gpurun_out/r2_checks
mkdir -p "$OUT"

run() {  # name timeout cmd...
  local name=$1 tmo=$2; shift 2
  echo "=== $name ==="
  timeout "$tmo" "$@" > "$OUT/$name.log" 2>&1
  echo "$name exit=$?" | tee -a "$OUT/summary.txt"
  tail -3 "$OUT/$name.log"
}

run regression_gpu 600 python -m pytest tests -m gpu -x -q
run gemm256 300 python scripts/validate_gemm256.py --bench
run attn 300 python scripts/validate_attention.py --bench
# native conv triple end-to-end through the autograd Function:
run conv_native 300 python - <<'EOF'
import json, torch, sys
sys.path.insert(0, ".")
import os
os.environ["TURBOPRUNE_CONV"] = "native"
from turboprune_amd.ops import conv_native
ok = True
for (cin, cout, k, s, hi) in [(64, 64, 3, 1, 56), (128, 128, 3, 2, 28),
                              (256, 64, 1, 1, 56), (64, 256, 1, 2, 56)]:
    torch.manual_seed(k)
    x = (torch.rand(16, cin, hi, hi, device="cuda") - .5).bfloat16() \
        .to(memory_format=torch.channels_last).requires_grad_()
    w = ((torch.rand(cout, cin, k, k, device="cuda") - .5) * .1).bfloat16() \
        .to(memory_format=torch.channels_last).requires_grad_()
    y = conv_native.conv2d(x, w, None, s, k // 2,
                           conv_native.NativeBackend)
    gy = torch.randn_like(y)
    gx, gw = torch.autograd.grad(y, [x, w], gy)
    # reference grads via autograd on F.conv2d
    x2 = x.detach().clone().requires_grad_(); w2 = w.detach().clone().requires_grad_()
    y2 = torch.nn.functional.conv2d(x2, w2, None, s, k // 2)
    y2.backward(gy)
    ey = (y.float() - y2.float()).abs().max().item()
    ex = (gx.float() - x2.grad.float()).abs().max().item()
    ew = (gw.float() - w2.grad.float()).abs().max().item()
    sc = max(y2.float().abs().max().item(), 1.0)
    good = ey < .05 * sc and ex < .5 and ew < .5
    ok &= good
    print(json.dumps({"shape": [cin, cout, k, s, hi], "ey": round(ey, 4),
                      "ex": round(ex, 4), "ew": round(ew, 4), "ok": good}))
print("PASS" if ok else "FAIL")
sys.exit(0 if ok else 1)
EOF
# same conv numerics with the 8-phase 256x256 variant routed in
# (shapes chosen to hit the Cout>=192 envelope incl. one at the edge):
run conv_native_256 300 env TURBOPRUNE_CONV256=1 python - <<'EOF'
import json, torch, sys, os
sys.path.insert(0, ".")
os.environ["TURBOPRUNE_CONV"] = "native"
from turboprune_amd.ops import conv_native
ok = True
for (cin, cout, k, s, hi) in [(64, 256, 3, 1, 28), (128, 256, 3, 2, 28),
                              (256, 512, 1, 2, 28), (64, 192, 1, 1, 28)]:
    torch.manual_seed(k + s)
    x = (torch.rand(16, cin, hi, hi, device="cuda") - .5).bfloat16() \
        .to(memory_format=torch.channels_last).requires_grad_()
    w = ((torch.rand(cout, cin, k, k, device="cuda") - .5) * .1).bfloat16() \
        .to(memory_format=torch.channels_last).requires_grad_()
    y = conv_native.conv2d(x, w, None, s, k // 2, conv_native.NativeBackend)
    gy = torch.randn_like(y)
    gx, gw = torch.autograd.grad(y, [x, w], gy)
    x2 = x.detach().clone().requires_grad_(); w2 = w.detach().clone().requires_grad_()
    y2 = torch.nn.functional.conv2d(x2, w2, None, s, k // 2)
    y2.backward(gy)
    ey = (y.float() - y2.float()).abs().max().item()
    ex = (gx.float() - x2.grad.float()).abs().max().item()
    ew = (gw.float() - w2.grad.float()).abs().max().item()
    sc = max(y2.float().abs().max().item(), 1.0)
    good = ey < .05 * sc and ex < .5 and ew < .5
    ok &= good
    print(json.dumps({"shape": [cin, cout, k, s, hi], "ey": round(ey, 4),
                      "ex": round(ex, 4), "ew": round(ew, 4), "ok": good}))
print("PASS" if ok else "FAIL")
sys.exit(0 if ok else 1)
EOF
# multi-tensor SGD parity vs per-tensor on a real model step:
run multi_sgd 300 python - <<'EOF'
import os, sys, torch
sys.path.insert(0, ".")
from turboprune_amd.config import compose
from turboprune_amd.models import build_model
from turboprune_amd.optim import FusedMaskedSGD
from turboprune_amd.ops import functional as TF

def one(env):
    os.environ.pop("TURBOPRUNE_MULTI_SGD", None)
    if env:
        os.environ["TURBOPRUNE_MULTI_SGD"] = "1"
    torch.manual_seed(0)
    cfg = compose("bench_resnet50_imagenet")
    pm = build_model(cfg).to("cuda").to(memory_format=torch.channels_last)
    pm.enable_caches(torch.bfloat16)
    opt = FusedMaskedSGD(pm.parameters(), lr=0.1, momentum=0.9,
                         weight_decay=1e-4, model=pm)
    x = torch.randn(8, 3, 224, 224, device="cuda") \
        .to(memory_format=torch.channels_last)
    y = torch.randint(0, 1000, (8,), device="cuda")
    for _ in range(3):
        opt.zero_grad(set_to_none=True)
        with torch.autocast("cuda", torch.bfloat16):
            loss = TF.cross_entropy(pm(x), y)
        loss.backward()
        opt.step()
    torch.cuda.synchronize()
    return {n: p.detach().clone() for n, p in pm.named_parameters()}

a = one(False)
b = one(True)
worst = max((a[n] - b[n]).abs().max().item() for n in a)
print({"multi_vs_single_max_diff": worst})
assert worst < 1e-5, worst
print("PASS")
EOF
# fused ScheduleFree step parity vs eager on a masked model:
run schedulefree 300 python - <<'EOF2'
import sys, torch
sys.path.insert(0, ".")
from turboprune_amd.ops.mask_layers import LinearMask
from turboprune_amd.optim import ScheduleFreeSGD

def one(native):
    import turboprune_amd.ops._backend as B
    torch.manual_seed(0)
    m = torch.nn.Sequential(LinearMask(in_features=64, out_features=32,
                                       bias=False)).cuda()
    m[0].mask.bernoulli_(0.5)
    m[0].enable_cache(torch.bfloat16)
    opt = ScheduleFreeSGD(m.parameters(), lr=0.05, momentum=0.9,
                          weight_decay=1e-3, model=m)
    opt.train()
    orig = B.use_native
    if not native:  # force eager by hiding the dispatch
        B.use_native = lambda *a, **k: False
    for _ in range(5):
        x = torch.randn(16, 64, device="cuda")
        loss = m(x).square().mean()
        opt.zero_grad(); loss.backward(); opt.step()
    B.use_native = orig
    torch.cuda.synchronize()
    return (m[0].weight.detach().clone(),
            m[0].weight_masked.detach().clone(),
            m[0].mask.detach().clone())

wf, cf, mask = one(True)
we, ce, _ = one(False)
d = (wf - we).abs().max().item()
dc = (cf.float() - (wf * mask).to(torch.bfloat16).float()).abs().max().item()
print({"fused_vs_eager_max_diff": d, "cache_vs_masked_w": dc})
assert d < 1e-5 and dc == 0.0, (d, dc)
print("PASS")
EOF2
# ---- promotion A/B benches (short; compare value fields) -------------
run bench_baseline 420 python bench.py --steps 15 --warmup 5
run bench_conv_native 420 env TURBOPRUNE_CONV=native \
    python bench.py --steps 15 --warmup 5
run bench_conv256 420 env TURBOPRUNE_CONV=native TURBOPRUNE_CONV256=1 \
    python bench.py --steps 15 --warmup 5
run bench_conv_wrwdb 420 env TURBOPRUNE_CONV=native TURBOPRUNE_WRW_DB=1 \
    python bench.py --steps 15 --warmup 5
run bench_multi_sgd 420 env TURBOPRUNE_MULTI_SGD=1 \
    python bench.py --steps 15 --warmup 5
run bench_deit_baseline 420 python bench.py --model deit_small \
    --global-batch 256 --steps 15 --warmup 5
run bench_deit_g256 420 env TURBOPRUNE_GEMM256=1 python bench.py \
    --model deit_small --global-batch 256 --steps 15 --warmup 5
run bench_deit_attn 420 env TURBOPRUNE_ATTN=native python bench.py \
    --model deit_small --global-batch 256 --steps 15 --warmup 5

echo "---- summary ----"
cat "$OUT/summary.txt"
grep -h '"value"' "$OUT"/bench_*.log 2>/dev/null | tail -12
