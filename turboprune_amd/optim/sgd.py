"""Optimizers.

``FusedMaskedSGD`` — SGD with momentum + weight decay whose GPU step is a
single HIP kernel per parameter fusing:

    momentum/wd update of the fp32 raw weight
    + rewrite of the bf16 *masked compute weight* cache (``mask ⊙ w``)

so the per-forward ``mask*weight`` multiply of the reference
(mask_layers.py:25) never runs in steady state (SURVEY K5 north-star).
Semantics match ``torch.optim.SGD(lr, momentum, weight_decay)`` exactly
(dampening 0, no nesterov): masked-out weights keep receiving wd/momentum
updates and are nullified only in the compute weight — so rewind
checkpoints contain the raw unmasked weights, as in the reference.

``ScheduleFreeSGD`` — a native implementation of Schedule-Free SGD
(Defazio et al., 2024; the reference imports the ``schedulefree`` package,
standard_pruning_harness.py:70-84): y-iterate in the params during train,
x-iterate swapped in for eval, z fast iterate in state.
"""

from __future__ import annotations

from typing import Any, Dict, Optional

import torch
import torch.nn as nn

from turboprune_amd.ops import _backend
from turboprune_amd.ops.mask_layers import masked_modules


class FusedMaskedSGD(torch.optim.SGD):
    """torch.optim.SGD subclass (state-dict compatible) with a fused
    HIP step for masked layers. Construct via ``build_optimizer`` or pass
    ``model`` so weight->(mask, layer) associations are discovered."""

    def __init__(self, params, lr: float, momentum: float = 0.0,
                 weight_decay: float = 0.0,
                 model: Optional[nn.Module] = None):
        super().__init__(params, lr=lr, momentum=momentum,
                         weight_decay=weight_decay)
        self._masked: Dict[int, Any] = {}
        if model is not None:
            self.attach_model(model)

    def attach_model(self, model: nn.Module) -> None:
        self._masked = {}
        for _, m in masked_modules(model):
            self._masked[id(m.weight)] = m

    @staticmethod
    def _multi_ok(p, grad, buf, mask, cache) -> bool:
        """All operands already flat in the weight's storage order — the
        multi-tensor kernel does no relayout (sgd_multi.hip contract)."""
        s = p.stride()
        for t in (grad, buf, mask, cache):
            if t is not None and (t.stride() != s or t.numel() != p.numel()):
                return False
        return p.dtype == torch.float32

    @torch.no_grad()
    def step(self, closure=None):  # noqa: C901
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        import os
        use_multi = os.environ.get("TURBOPRUNE_MULTI_SGD", "") == "1"

        for group in self.param_groups:
            lr = group["lr"]
            momentum = group["momentum"]
            wd = group["weight_decay"]
            buckets: Dict[Any, list] = {}
            for p in group["params"]:
                if p.grad is None:
                    continue
                grad = p.grad
                state = self.state[p]
                if momentum != 0:
                    buf = state.get("momentum_buffer")
                    if buf is None:
                        # zero-init ≡ torch's first-step clone at dampening 0
                        buf = torch.zeros_like(p)
                        state["momentum_buffer"] = buf
                else:
                    buf = None

                layer = self._masked.get(id(p))
                cache = getattr(layer, "weight_masked", None) \
                    if layer is not None else None
                mask = layer.mask if layer is not None else None

                if p.is_cuda and _backend.use_native(p):
                    ext = _backend.extension()
                    if use_multi and self._multi_ok(p, grad, buf, mask,
                                                    cache):
                        key = (grad.dtype, mask is not None,
                               cache is not None,
                               cache.dtype if cache is not None else None)
                        buckets.setdefault(key, []).append(
                            (p, grad, buf, mask, cache))
                        continue
                    ext.sgd_step_(
                        p, grad, buf if buf is not None else torch.Tensor(),
                        mask if mask is not None else torch.Tensor(),
                        cache if cache is not None else torch.Tensor(),
                        float(lr), float(momentum), float(wd))
                else:
                    d_p = grad.to(p.dtype)
                    if wd != 0:
                        d_p = d_p.add(p, alpha=wd)
                    if buf is not None:
                        buf.mul_(momentum).add_(d_p)
                        d_p = buf
                    p.add_(d_p, alpha=-lr)
                    if cache is not None and layer is not None:
                        layer.refresh_cache()
            for (gdt, has_mask, has_cache, cdt), items in buckets.items():
                ext = _backend.extension()
                ws = [it[0] for it in items]
                gs = [it[1] for it in items]
                bufs = [it[2] for it in items] if momentum != 0 else []
                masks = [it[3] for it in items] if has_mask else []
                caches = [it[4] for it in items] if has_cache else []
                # plan cache keyed by every operand's data_ptr: pointers
                # are stable in steady state (params/bufs/masks fixed,
                # allocator reuses grad blocks); ANY reallocation changes
                # the key and rebuilds — a stale plan cannot be hit.
                key = tuple(t.data_ptr()
                            for grp in (ws, gs, bufs, masks, caches)
                            for t in grp)
                plans = getattr(self, "_multi_plans", None)
                if plans is None:
                    plans = self._multi_plans = {}
                plan = plans.get(key)
                if plan is None:
                    if len(plans) > 64:  # ptr churn: don't grow unbounded
                        plans.clear()
                    plan = plans[key] = ext.sgd_multi_plan(
                        ws, gs, bufs, masks, caches)
                ext.sgd_step_multi_planned_(
                    plan[0], plan[1], momentum != 0, has_mask, has_cache,
                    gdt == torch.bfloat16, cdt == torch.bfloat16,
                    float(lr), float(momentum), float(wd))
        return loss


class ScheduleFreeSGD(torch.optim.Optimizer):
    """Schedule-Free SGD (y/z/x iterates). ``train()`` must be called
    before training steps and ``eval()`` before evaluation — params hold
    the y iterate while training and the x (Polyak-averaged) iterate for
    eval, as in the schedulefree package the reference uses."""

    def __init__(self, params, lr: float, momentum: float = 0.9,
                 weight_decay: float = 0.0, warmup_steps: int = 0,
                 model: Optional[nn.Module] = None):
        defaults = dict(lr=lr, momentum=momentum, weight_decay=weight_decay,
                        warmup_steps=warmup_steps, k=0, train_mode=False,
                        weight_sum=0.0)
        super().__init__(params, defaults)
        self._masked: Dict[int, Any] = {}
        if model is not None:
            self.attach_model(model)

    def attach_model(self, model: nn.Module) -> None:
        """weight->masked-layer map so the fused GPU step can rewrite
        the bf16 masked-cache in the same sweep (without it the eager
        y-iterate mutations bump every weight's _version and force a
        mask_apply per layer per forward)."""
        self._masked = {}
        for _, m in masked_modules(model):
            self._masked[id(m.weight)] = m

    @torch.no_grad()
    def train(self):
        for group in self.param_groups:
            if not group["train_mode"]:
                beta = group["momentum"]
                for p in group["params"]:
                    st = self.state.get(p)
                    if st and "z" in st:
                        # x -> y = (1-beta)*z + beta*x
                        p.lerp_(st["z"], weight=1.0 - beta)
                group["train_mode"] = True

    @torch.no_grad()
    def eval(self):
        for group in self.param_groups:
            if group["train_mode"]:
                beta = group["momentum"]
                for p in group["params"]:
                    st = self.state.get(p)
                    if st and "z" in st:
                        # y -> x = (y - (1-beta)*z) / beta
                        p.sub_(st["z"], alpha=1.0 - beta).div_(beta)
                group["train_mode"] = False

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group in self.param_groups:
            if not group["train_mode"]:
                raise RuntimeError("ScheduleFreeSGD.step() called in eval "
                                   "mode; call optimizer.train() first")
            k = group["k"]
            warmup = group["warmup_steps"]
            sched = (k + 1) / warmup if k < warmup else 1.0
            lr = group["lr"] * sched
            beta = group["momentum"]
            wd = group["weight_decay"]

            lr_max = group["lr_max"] = max(lr, group.get("lr_max", 0.0))
            weight = lr_max ** 2
            weight_sum = group["weight_sum"] = group["weight_sum"] + weight
            ckp1 = weight / weight_sum if weight_sum else 0.0

            for p in group["params"]:
                if p.grad is None:
                    continue
                grad = p.grad
                st = self.state[p]
                if "z" not in st:
                    st["z"] = p.detach().clone()
                z = st["z"]
                layer = self._masked.get(id(p))
                cache = getattr(layer, "weight_masked", None) \
                    if layer is not None else None
                if (p.is_cuda and _backend.use_native(p)
                        and grad.stride() == p.stride()
                        and p.dtype == torch.float32):
                    _backend.extension().schedulefree_step_(
                        p, z, grad,
                        layer.mask if layer is not None else torch.Tensor(),
                        cache if cache is not None else torch.Tensor(),
                        float(lr), float(beta), float(ckp1), float(wd))
                    continue
                if wd != 0:
                    grad = grad.add(p, alpha=wd)
                # y -> x step then x -> y with new z
                p.lerp_(z, weight=ckp1)
                p.add_(grad, alpha=lr * (beta * (1.0 - ckp1) - 1.0))
                z.sub_(grad, alpha=lr)
            group["k"] = k + 1
        return loss


def build_optimizer(cfg: Any, model: nn.Module) -> torch.optim.Optimizer:
    """Optimizer from config (reference:
    standard_pruning_harness.py:52-84: SGD or schedulefree)."""
    op = cfg.optimizer_params
    lr = float(op.lr)
    momentum = float(op.momentum)
    wd = float(op.weight_decay)
    if op.scheduler_type == "ScheduleFree":
        warmup = int(cfg.select("optimizer_params.warmup_steps", 0))
        return ScheduleFreeSGD(model.parameters(), lr=lr, momentum=momentum,
                               weight_decay=wd, warmup_steps=warmup,
                               model=model)
    return FusedMaskedSGD(model.parameters(), lr=lr, momentum=momentum,
                          weight_decay=wd, model=model)
