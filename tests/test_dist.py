"""Multi-process (gloo, world_size=2) tests for the first-party DP engine and
metric sync — the CPU stand-in for the RCCL/xGMI path."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from cilfw.models import CilModel
from cilfw.distributed.ddp import DataParallelEngine
from cilfw.optim import FlatSGD
from cilfw.utils.metrics import MetricLogger
from cilfw import ops


def _init(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)


def _run_engine_worker(rank, world, port, out):
    _init(rank, world, port)
    torch.manual_seed(42)  # same init everywhere; broadcast also enforces it
    model = CilModel("resnet20", 32)
    model.prev_model_adaption(4)
    engine = DataParallelEngine(model, bucket_mb=0.05)  # force several buckets
    opt = FlatSGD(engine, lr=0.1, momentum=0.9, weight_decay=0.0)

    # global batch of 8: rank r takes rows r*4:(r+1)*4
    g = torch.Generator().manual_seed(7)
    x = torch.randn(8, 16, 16, 3, generator=g)
    y = torch.randint(0, 4, (8,), generator=g)
    xs = x[rank * 4:(rank + 1) * 4]
    ys = y[rank * 4:(rank + 1) * 4]

    opt.zero_grad()
    logits, _ = model(xs)
    loss = ops.cross_entropy(logits.float(), ys)
    loss.backward()
    engine.finalize()
    grad = engine.flat_grads.clone()
    opt.step()
    out[rank] = {"grad": grad, "params": engine.flat_params.clone()}
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_dp_engine_matches_single_process():
    world = 2
    port = 29611
    mgr = mp.Manager()
    out = mgr.dict()
    mp.spawn(_run_engine_worker, args=(world, port, out), nprocs=world,
             join=True)

    # single-process oracle: mean of the two HALF-batch grads (BN statistics are
    # per-rank — reference semantics, SURVEY §2.3 K3 — so a full-batch run is
    # NOT equivalent; the engine must equal the average of per-shard grads).
    g = torch.Generator().manual_seed(7)
    x = torch.randn(8, 16, 16, 3, generator=g)
    y = torch.randint(0, 4, (8,), generator=g)
    shard_grads = []
    for r in range(world):
        torch.manual_seed(42)
        model = CilModel("resnet20", 32)
        model.prev_model_adaption(4)
        engine = DataParallelEngine(model, bucket_mb=0.05)
        engine.zero_grad()
        logits, _ = model(x[r * 4:(r + 1) * 4])
        loss = ops.cross_entropy(logits.float(), y[r * 4:(r + 1) * 4])
        loss.backward()
        engine.finalize()
        shard_grads.append(engine.flat_grads.clone())
    oracle = (shard_grads[0] + shard_grads[1]) / 2

    assert torch.allclose(out[0]["grad"], out[1]["grad"], atol=1e-6), \
        "ranks must end with identical averaged gradients"
    assert torch.allclose(out[0]["grad"], oracle, atol=1e-5)
    assert torch.allclose(out[0]["params"], out[1]["params"], atol=1e-6)


def _run_metric_worker(rank, world, port, out):
    _init(rank, world, port)
    ml = MetricLogger()
    # rank 0: 3 samples of value 1; rank 1: 1 sample of value 5
    if rank == 0:
        ml.update_n(n=3, acc=1.0)
    else:
        ml.update_n(n=1, acc=5.0)
    ml.synchronize_between_processes(device=torch.device("cpu"))
    out[rank] = ml.meters["acc"].global_avg
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_metric_sync_exact_weighted_mean():
    mgr = mp.Manager()
    out = mgr.dict()
    mp.spawn(_run_metric_worker, args=(2, 29613, out), nprocs=2, join=True)
    # (3*1 + 1*5) / 4 = 2.0 on every rank
    assert abs(out[0] - 2.0) < 1e-9
    assert abs(out[1] - 2.0) < 1e-9


def _run_broadcast_worker(rank, world, port, out):
    _init(rank, world, port)
    torch.manual_seed(100 + rank)  # DIFFERENT init per rank
    model = CilModel("resnet20", 32)
    model.prev_model_adaption(4)
    engine = DataParallelEngine(model)
    out[rank] = engine.flat_params.clone()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_engine_broadcasts_rank0_weights():
    mgr = mp.Manager()
    out = mgr.dict()
    mp.spawn(_run_broadcast_worker, args=(2, 29615, out), nprocs=2, join=True)
    assert torch.equal(out[0], out[1])


def _run_bucket_fuzz_worker(rank, world, port, out, bucket_mb):
    _init(rank, world, port)
    torch.manual_seed(42)
    model = CilModel("resnet20", 32)
    model.prev_model_adaption(4)
    engine = DataParallelEngine(model, bucket_mb=bucket_mb)
    opt = FlatSGD(engine, lr=0.1, momentum=0.9, weight_decay=0.0)
    g = torch.Generator().manual_seed(7)
    x = torch.randn(8, 16, 16, 3, generator=g)
    y = torch.randint(0, 4, (8,), generator=g)
    opt.zero_grad()
    logits, _ = model(x[rank * 4:(rank + 1) * 4])
    loss = ops.cross_entropy(logits.float(), y[rank * 4:(rank + 1) * 4])
    loss.backward()
    engine.finalize()
    out[(rank, bucket_mb)] = engine.flat_grads.clone()
    dist.destroy_process_group()


@pytest.mark.timeout(600)
def test_bucket_boundary_fuzz():
    """Averaged gradients must be identical for ANY bucket size — one huge
    bucket, many tiny ones, and sizes that cut mid-layer."""
    mgr = mp.Manager()
    out = mgr.dict()
    sizes = [100.0, 0.5, 0.037, 0.011, 0.003]
    for i, mb in enumerate(sizes):
        mp.spawn(_run_bucket_fuzz_worker, args=(2, 29661 + i, out, mb),
                 nprocs=2, join=True)
    ref = out[(0, sizes[0])]
    for mb in sizes:
        assert torch.allclose(out[(0, mb)], out[(1, mb)], atol=1e-6), mb
        assert torch.allclose(out[(0, mb)], ref, atol=1e-6), mb


def _run_nosync_worker(rank, world, port, out):
    _init(rank, world, port)
    torch.manual_seed(42)
    model = CilModel("resnet20", 32)
    model.prev_model_adaption(4)
    engine = DataParallelEngine(model, bucket_mb=0.05)
    opt = FlatSGD(engine, lr=0.1, momentum=0.9, weight_decay=0.0)
    g = torch.Generator().manual_seed(9)
    xs = [torch.randn(8, 16, 16, 3, generator=g) for _ in range(2)]
    ys = [torch.randint(0, 4, (8,), generator=g) for _ in range(2)]
    # 2 accumulation micro-batches under no_sync, then a synced one
    opt.zero_grad()
    with engine.no_sync():
        for x, y in zip(xs, ys):
            logits, _ = model(x[rank * 4:(rank + 1) * 4])
            ops.cross_entropy(logits.float(),
                              y[rank * 4:(rank + 1) * 4]).backward()
    accum = engine.flat_grads.clone()
    engine.finalize()  # flushes + averages the accumulated grads
    out[rank] = {"accum": accum, "avg": engine.flat_grads.clone()}
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_no_sync_accumulates_then_syncs():
    """Grad accumulation: per-rank sums under no_sync (sink accumulate mode),
    then one finalize averages across ranks."""
    mgr = mp.Manager()
    out = mgr.dict()
    mp.spawn(_run_nosync_worker, args=(2, 29671, out), nprocs=2, join=True)
    # pre-finalize grads are rank-local sums => DIFFERENT across ranks
    assert not torch.allclose(out[0]["accum"], out[1]["accum"], atol=1e-6)
    # post-finalize: identical averages
    assert torch.allclose(out[0]["avg"], out[1]["avg"], atol=1e-6)
    expect = (out[0]["accum"] + out[1]["accum"]) / 2
    assert torch.allclose(out[0]["avg"], expect, atol=1e-5)


def _run_uneven_eval_worker(rank, world, port, out):
    _init(rank, world, port)
    from cilfw.data.sampler import DistributedSampler

    class _DS:
        def __len__(self):
            return 7  # NOT divisible by world=2 -> pad-by-repetition

    s = DistributedSampler(_DS(), world, rank, shuffle=False)
    out[rank] = list(iter(s))
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_uneven_shard_padding_across_ranks():
    mgr = mp.Manager()
    out = mgr.dict()
    mp.spawn(_run_uneven_eval_worker, args=(2, 29681, out), nprocs=2,
             join=True)
    allidx = sorted(out[0] + out[1])
    assert len(out[0]) == len(out[1]) == 4  # equal shards
    assert set(allidx) == set(range(7))     # every sample covered
    assert allidx == [0, 0, 1, 2, 3, 4, 5, 6]  # torch pad-from-front
