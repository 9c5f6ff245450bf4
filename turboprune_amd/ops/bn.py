"""Fused BatchNorm(+ReLU)(+residual-add) — the ResNet epilogue path.

Profiling the ResNet50 bench on MI355X showed MIOpen BatchNorm + eager
ReLU + eager residual add at ~60% of step GPU time
(profiles/r01_bench_resnet50_1gpu_baseline.md). ``bn_act`` runs the whole
BN->(+res)->ReLU epilogue as 3 HIP kernels forward (reduce / finalize /
apply) and 3 backward, NHWC-coalesced, replacing ~10 library/eager
launches per block.

Semantics match ``nn.BatchNorm2d`` exactly (biased batch var for
normalization, unbiased for running_var, momentum update,
num_batches_tracked). The composed-torch implementation below is the CPU
path and the numerics oracle; the fused path engages on channels_last
GPU tensors in train mode (and in eval under no_grad).
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from turboprune_amd.ops import _backend


class _FusedBN(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, residual, gamma, beta, running_mean, running_var,
                training, momentum, eps, relu):
        ext = _backend.extension()
        y, mean, rstd, rmask = ext.bn_fwd(x, residual, gamma, beta,
                                          running_mean, running_var,
                                          training, momentum, eps, relu)
        ctx.save_for_backward(x, gamma, mean, rstd, rmask)
        ctx.relu = relu
        ctx.has_res = residual is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        x, gamma, mean, rstd, rmask = ctx.saved_tensors
        ext = _backend.extension()
        if rmask is None:  # relu=False path: no bitmask was written
            rmask = torch.Tensor()
        dx, dgamma, dbeta, dres = ext.bn_bwd(x, rmask, dy, gamma, mean,
                                             rstd, ctx.relu, ctx.has_res)
        return (dx, dres if ctx.has_res else None,
                dgamma.to(gamma.dtype), dbeta.to(gamma.dtype),
                None, None, None, None, None, None)


def _composed_bn_act(bn: nn.BatchNorm2d, x, residual, relu,
                     momentum: float):
    """Reference implementation (the oracle / CPU path)."""
    y = F.batch_norm(x, bn.running_mean, bn.running_var, bn.weight,
                     bn.bias, bn.training, momentum, bn.eps)
    if residual is not None:
        y = y + residual
    if relu:
        y = F.relu(y)
    return y


def bn_act(bn: nn.BatchNorm2d, x: torch.Tensor,
           residual: Optional[torch.Tensor] = None,
           relu: bool = False) -> torch.Tensor:
    """BatchNorm through the module's params/buffers, with optional fused
    residual add and ReLU."""
    use_fused = (
        x.is_cuda
        and x.dim() == 4
        and x.is_contiguous(memory_format=torch.channels_last)
        and x.dtype in (torch.bfloat16, torch.float32)
        and (residual is None or residual.dtype == x.dtype)
        and bn.track_running_stats
        and (bn.training or not torch.is_grad_enabled())
        and _backend.use_native(x)
    )
    if use_fused:
        use_fused = _backend.extension().bn_fast_path_ok(x)
    if bn.training and bn.track_running_stats \
            and bn.num_batches_tracked is not None:
        bn.num_batches_tracked.add_(1)
    # effective momentum handles momentum=None on BOTH paths (F.batch_norm
    # raises on None)
    momentum = bn.momentum if bn.momentum is not None \
        else (1.0 / float(bn.num_batches_tracked)
              if bn.num_batches_tracked is not None
              and int(bn.num_batches_tracked) > 0 else 0.1)
    if not use_fused:
        return _composed_bn_act(bn, x, residual, relu, momentum)
    if bn.training and torch.is_grad_enabled():
        return _FusedBN.apply(x, residual, bn.weight, bn.bias,
                              bn.running_mean, bn.running_var, True,
                              momentum, bn.eps, relu)
    ext = _backend.extension()
    with torch.no_grad():
        y = ext.bn_fwd(x, residual, bn.weight, bn.bias,
                       bn.running_mean, bn.running_var, bn.training,
                       momentum, bn.eps, relu)[0]
    return y


class FusedBatchNorm2d(nn.BatchNorm2d):
    """Drop-in nn.BatchNorm2d whose plain forward routes through the
    fused kernel (no ReLU); blocks call ``bn_act`` directly for the
    ReLU/residual fusions. State dict identical to nn.BatchNorm2d."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return bn_act(self, x, residual=None, relu=False)
