"""Full CIL run under gloo world_size=2 — the multi-process end-to-end path
(sampler sharding, DP engine, metric sync, rank-identical memory/WA)."""

import os

import pytest
import torch.distributed as dist
import torch.multiprocessing as mp

from cilfw.config import parse_args
from cilfw.engine import run


def _worker(rank, world, port, out):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["LOCAL_RANK"] = str(rank)
    args = parse_args([
        "--data_set", "synthetic", "--backbone", "resnet20",
        "--synthetic_classes", "10", "--num_bases", "5", "--increment", "5",
        "--num_epochs", "2", "--batch_size", "16", "--workers", "0",
        "--synthetic_train_size", "400", "--memory_size", "20",
        "--eval_every_epoch", "0", "--input_size", "16", "--no_aug",
        "--lr", "0.05", "--seed", "5",
    ])
    accs = run(args)
    out[rank] = accs
    if dist.is_initialized():
        dist.destroy_process_group()


@pytest.mark.timeout(900)
def test_distributed_cil_run_ws2():
    mgr = mp.Manager()
    out = mgr.dict()
    mp.spawn(_worker, args=(2, 29721, out), nprocs=2, join=True)
    assert len(out) == 2
    # every rank computes the same (globally synchronized) accuracies
    assert out[0] == out[1]
    assert len(out[0]) == 2
