"""RCCL collectives exercised on hardware: two
ranks co-located on the single leased MI355X, torch.distributed backend
"nccl" (= RCCL on ROCm). Covers the broadcast -> equality-check ->
DDP-grad-all-reduce path that the gloo CPU tests cover, but over the
real collective library the 8-GPU runs will use."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

pytestmark = pytest.mark.gpu


def _worker(rank, world, port, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        # both ranks share cuda:0 — RCCL supports co-located ranks
        dist.init_process_group("nccl", rank=rank, world_size=world)
        torch.cuda.set_device(0)
        from turboprune_amd.ops.mask_layers import LinearMask
        from turboprune_amd.parallel.ddp import (broadcast_model_state,
                                                 check_model_equality)
        torch.manual_seed(rank)  # deliberately different per rank
        model = torch.nn.Sequential(
            LinearMask(in_features=64, out_features=32),
            torch.nn.ReLU(),
            LinearMask(in_features=32, out_features=8),
        ).to("cuda:0")
        if rank == 0:
            with torch.no_grad():
                model[0].mask.bernoulli_(0.5)
        broadcast_model_state(model, src=0)
        eq = check_model_equality(model)

        # DDP grad all-reduce parity: average of per-rank grads
        ddp = torch.nn.parallel.DistributedDataParallel(
            model, broadcast_buffers=False)
        torch.manual_seed(1234)
        x = torch.randn(2 * world, 64, device="cuda:0")
        y = torch.randn(2 * world, 8, device="cuda:0")
        loss = torch.nn.functional.mse_loss(
            ddp(x[rank * 2:(rank + 1) * 2]), y[rank * 2:(rank + 1) * 2])
        loss.backward()
        g = model[0].weight.grad.detach().cpu().tolist()
        torch.cuda.synchronize()
        q.put(("ok", rank, eq, g))
        dist.destroy_process_group()
    except BaseException as e:  # noqa: BLE001
        q.put(("err", rank, repr(e), None))
        raise


@pytest.mark.timeout(300)
def test_rccl_two_ranks_one_gpu():
    world = 2
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29611
    procs = [ctx.Process(target=_worker, args=(r, world, port, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = [q.get(timeout=240) for _ in range(world)]
    for p in procs:
        p.join(timeout=60)
    errs = [r for r in results if r[0] == "err"]
    assert not errs, errs
    assert all(r[2] for r in results), "post-broadcast equality failed"
    g0, g1 = (torch.tensor(r[3]) for r in sorted(results,
                                                 key=lambda r: r[1]))
    # all-reduced grads identical across ranks
    assert torch.allclose(g0, g1, atol=1e-6)


def _allreduce_worker(rank, world, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = "29612"
        dist.init_process_group("nccl", rank=rank, world_size=world)
        torch.cuda.set_device(0)
        for numel in (1 << 10, 1 << 20, 8 << 20):
            t = torch.full((numel,), float(rank + 1), device="cuda:0")
            dist.all_reduce(t)
            assert t[0].item() == 3.0 and t[-1].item() == 3.0
        torch.cuda.synchronize()
        q.put(("ok", rank))
        dist.destroy_process_group()
    except BaseException as e:  # noqa: BLE001
        q.put(("err", rank, repr(e)))
        raise


@pytest.mark.timeout(300)
def test_rccl_allreduce_bucket_sizes():
    """Plain all_reduce over RCCL at the DDP bucket sizes we ship
    (parallel/ddp.py): correctness of the collective itself."""
    world = 2
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_allreduce_worker, args=(r, world, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = [q.get(timeout=240) for _ in range(world)]
    for p in procs:
        p.join(timeout=60)
    assert all(r[0] == "ok" for r in results), results
