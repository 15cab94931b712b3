"""Multi-process DP correctness on CPU (gloo): gradients from
BucketedDataParallel across 2 ranks must equal the single-process
gradients on the combined batch."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from deepof_amd.models import FlowNetS


def _make_model(seed=0):
    torch.manual_seed(seed)
    return FlowNetS()


def _loss(model, x):
    flows = model(x)
    return sum(f.float().pow(2).mean() for f in flows)


def _worker(rank, world, port, results):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["RANK"] = str(rank)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from deepof_amd.parallel import BucketedDataParallel

        base = _make_model().to(memory_format=torch.channels_last)
        model = BucketedDataParallel(base, bucket_cap_mb=4)
        torch.manual_seed(100 + rank)
        x = torch.randn(2, 6, 32, 48).to(memory_format=torch.channels_last)
        loss = _loss(model, x)
        loss.backward()
        model.finish_gradient_sync()
        named = [(n, p) for n, p in model.module.named_parameters()
                 if p.grad is not None]
        keep = named[:3] + named[-3:]  # keep the manager-dict payload small
        results[rank] = {n: p.grad.clone() for n, p in keep}
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_bucketed_ddp_matches_single_process():
    world = 2
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        results = mgr.dict()
        port = 29511
        procs = [ctx.Process(target=_worker, args=(r, world, port, results))
                 for r in range(world)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(240)
            assert p.exitcode == 0
        ddp_grads = {k: v for k, v in results[0].items()}

    # single-process on the combined batch: average of per-rank losses
    model = _make_model().to(memory_format=torch.channels_last)
    xs = []
    for rank in range(world):
        torch.manual_seed(100 + rank)
        xs.append(torch.randn(2, 6, 32, 48).to(
            memory_format=torch.channels_last))
    loss = sum(_loss(model, x) for x in xs) / world
    loss.backward()

    single = dict(model.named_parameters())
    assert ddp_grads
    for n, g in ddp_grads.items():
        torch.testing.assert_close(g, single[n].grad, rtol=1e-4, atol=1e-6)


def _accum_worker(rank, world, port, results):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["RANK"] = str(rank)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from deepof_amd.parallel import BucketedDataParallel

        base = _make_model().to(memory_format=torch.channels_last)
        model = BucketedDataParallel(base, bucket_cap_mb=4)
        torch.manual_seed(100 + rank)
        xa = torch.randn(2, 6, 32, 48).to(memory_format=torch.channels_last)
        xb = torch.randn(2, 6, 32, 48).to(memory_format=torch.channels_last)

        # micro-batch 1: accumulate only (re-arm, no all-reduce)
        model.accumulate_only = True
        _loss(model, xa).backward()
        # micro-batch 2: boundary -> reduce the accumulated sums
        model.accumulate_only = False
        _loss(model, xb).backward()
        model.finish_gradient_sync()
        named = [(n, p) for n, p in model.module.named_parameters()
                 if p.grad is not None]
        keep = named[:3] + named[-3:]
        results[rank] = {n: p.grad.clone() for n, p in keep}
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_bucketed_ddp_gradient_accumulation():
    """accumulate_only micro-batches must sum locally and reduce once:
    final grads == average over ranks of (grad(xa) + grad(xb))."""
    world = 2
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        results = mgr.dict()
        procs = [ctx.Process(target=_accum_worker,
                             args=(r, world, 29517, results))
                 for r in range(world)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(240)
            assert p.exitcode == 0
        got = dict(results[0])

    # single-process reference: mean over the 4 micro-batch losses' sums
    ref_model = _make_model().to(memory_format=torch.channels_last)
    for r in range(world):
        torch.manual_seed(100 + r)
        xa = torch.randn(2, 6, 32, 48).to(memory_format=torch.channels_last)
        xb = torch.randn(2, 6, 32, 48).to(memory_format=torch.channels_last)
        _loss(ref_model, xa).backward()
        _loss(ref_model, xb).backward()
    ref = dict(ref_model.named_parameters())
    for n, g in got.items():
        want = ref[n].grad / world
        torch.testing.assert_close(g, want, rtol=1e-4, atol=1e-6)
