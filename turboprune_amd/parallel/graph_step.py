"""hipGraph-captured train step with a flat-buffer gradient all-reduce —
the small-per-rank-batch (strong-scaling) path.

At global batch 512 on 8 GPUs the per-rank bs-64 step is launch-bound
(~500 host launches for ~16 ms of GPU work).
torch DDP's hook-driven bucketing cannot be captured directly, so this
module replaces it for the steady-state training step:

- every parameter's ``.grad`` is re-pointed at a VIEW into one
  contiguous fp32 buffer (autograd accumulates into existing .grad
  tensors when ``zero_grad(set_to_none=False)`` — addresses stay fixed,
  which is exactly what graph capture needs);
- the captured graph is: zero flat buffer -> autocast fwd -> CE ->
  backward -> ONE ``all_reduce(flat, AVG)`` over RCCL -> fused SGD step.
  RCCL collectives are capturable once the communicator is warm (torch
  ProcessGroupNCCL supports stream capture);
- replay does: copy batch into static input buffers -> graph.replay().

Trade-off vs DDP: the all-reduce is not overlapped with backward — one
~102 MB fp32 ring all-reduce over 7 xGMI links costs ~1-2 ms at N=8,
far less than the launch overhead the replay removes. At large per-rank
batches DDP's overlap wins, so ``wanted()`` gates on per-rank batch.

The flat-grad step (without capture) is CPU-testable under gloo:
``FlatGradStep`` alone must match DDP gradients exactly
(tests/test_distributed.py).
"""

from __future__ import annotations

from typing import Callable, Optional

import torch
import torch.distributed as dist


class FlatGradStep:
    """Grads as views into one flat fp32 buffer + a single averaged
    all-reduce. Works on CPU (gloo) and GPU; capture-friendly."""

    def __init__(self, params, process_group=None):
        self.params = [p for p in params if p.requires_grad]
        assert all(p.dtype == torch.float32 for p in self.params), \
            "flat grad buffer assumes fp32 master params"
        total = sum(p.numel() for p in self.params)
        dev = self.params[0].device
        self.flat = torch.zeros(total, dtype=torch.float32, device=dev)
        off = 0
        for p in self.params:
            n = p.numel()
            g = self.flat[off:off + n].view_as(p)
            if p.is_contiguous(memory_format=torch.channels_last):
                # autograd produces channels_last grads for these; the
                # view must match the grad layout for in-place accumulate
                g = self.flat[off:off + n].view(
                    p.permute(0, 2, 3, 1).shape).permute(0, 3, 1, 2)
            p.grad = g
            off += n
        self.group = process_group

    def zero_(self):
        self.flat.zero_()

    def allreduce_(self):
        if dist.is_available() and dist.is_initialized():
            world = dist.get_world_size(self.group)
            if world > 1:
                dist.all_reduce(self.flat, group=self.group)
                self.flat.div_(world)


class GraphedTrainStep:
    """Capture (zero -> fwd -> loss -> bwd -> all-reduce -> opt.step) in
    one hipGraph; replay per batch. Falls back is the caller's job (see
    bench.py): construction raises if capture fails."""

    def __init__(self, model, opt, loss_fn: Callable, example_x,
                 example_y, bf16: bool = True, warmup_iters: int = 3):
        self.static_x = example_x.clone()
        self.static_y = example_y.clone()
        self.flat = FlatGradStep([p for g in opt.param_groups
                                  for p in g["params"]])

        def body():
            self.flat.zero_()
            with torch.autocast(device_type="cuda", dtype=torch.bfloat16,
                                enabled=bf16):
                out = model(self.static_x)
                loss = loss_fn(out, self.static_y)
            loss.backward()
            self.flat.allreduce_()
            opt.step()
            return loss

        # warmup on a side stream (capture requirement; also warms the
        # RCCL communicator and the allocator into steady state)
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(warmup_iters):
                last = body()
        torch.cuda.current_stream().wait_stream(side)
        del last
        torch.cuda.synchronize()

        # autograd must have accumulated IN PLACE into the flat views;
        # a rebound .grad would make the captured all-reduce silently
        # reduce stale memory
        lo = self.flat.flat.data_ptr()
        hi = lo + self.flat.flat.numel() * 4
        for p in self.flat.params:
            assert p.grad is not None and lo <= p.grad.data_ptr() < hi, \
                "a grad escaped the flat buffer; cannot capture"

        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            self.loss = body()

    def __call__(self, x, y):
        self.static_x.copy_(x)
        self.static_y.copy_(y)
        self.graph.replay()
        return self.loss


def wanted(per_rank_batch: int, distributed: bool) -> bool:
    """Auto-enable heuristic — currently OFF by measurement: at bs64 on
    MI355X the captured replay matched eager within noise (16.9 vs
    16.5 ms, r2d) because the small-batch step is GPU-busy, not
    launch-bound, and the captured flat all-reduce gives up DDP's
    backward overlap. Capture stays available via --graph; flip this
    back if a future kernel generation makes the step launch-bound."""
    return False
