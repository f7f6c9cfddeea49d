import numpy as np
import torch

from cilfw.data.gpu_pipeline import GpuTaskLoader
from cilfw.data.scenario import TaskSet
from cilfw.data.sampler import DistributedSampler


def _taskset(n=50, size=8):
    rng = np.random.default_rng(0)
    x = rng.integers(0, 255, (n, size, size, 3), dtype=np.uint8)
    y = rng.integers(0, 5, n).astype(np.int64)
    return TaskSet(x, y, np.zeros(n, dtype=np.int64))


def test_batch_shapes_and_normalization():
    ts = _taskset()
    ld = GpuTaskLoader(ts, 16, "cpu", (0.5, 0.5, 0.5), (0.25, 0.25, 0.25),
                       augment=False, shuffle=False, dtype=torch.float32)
    batches = list(ld)
    assert len(batches) == 3  # drop_last: 50 // 16
    imgs, labels, _ = batches[0]
    assert imgs.shape == (16, 8, 8, 3) and labels.shape == (16,)
    expect = (torch.from_numpy(ts.x[:16]).float() - 0.5 * 255) / (0.25 * 255)
    assert torch.allclose(imgs, expect, atol=1e-5)


def test_sharding_matches_distributed_sampler():
    ts = _taskset(n=53)
    for world in (1, 2):
        for rank in range(world):
            ld = GpuTaskLoader(ts, 8, "cpu", (0.5,) * 3, (0.25,) * 3,
                               world=world, rank=rank, shuffle=True, seed=9,
                               augment=False, drop_last=False,
                               dtype=torch.float32)
            ld.set_epoch(2)
            ref = DistributedSampler(ts, world, rank, shuffle=True, seed=9)
            ref.set_epoch(2)
            got = torch.cat([lab for _, lab, _ in ld])
            want = torch.tensor([int(ts.y[i]) for i in ref])
            assert torch.equal(got, want)


def test_augment_preserves_stats():
    ts = _taskset(n=64, size=16)
    ld = GpuTaskLoader(ts, 64, "cpu", (0.5,) * 3, (0.25,) * 3, augment=True,
                       shuffle=False, dtype=torch.float32)
    imgs, _, _ = next(iter(ld))
    assert imgs.shape == (64, 16, 16, 3)
    assert torch.isfinite(imgs).all()


def test_drop_last_and_epoch_reshuffle():
    ts = _taskset(n=50)
    ld = GpuTaskLoader(ts, 16, "cpu", (0.5,) * 3, (0.25,) * 3, shuffle=True,
                       seed=3, augment=False, drop_last=True,
                       dtype=torch.float32)
    ld.set_epoch(0)
    e0 = torch.cat([lab for _, lab, _ in ld])
    ld.set_epoch(1)
    e1 = torch.cat([lab for _, lab, _ in ld])
    assert len(e0) == 48 and len(e1) == 48  # 3 batches, drop_last
    assert not torch.equal(e0, e1)  # epoch changes the permutation
    ld.set_epoch(0)
    e0b = torch.cat([lab for _, lab, _ in ld])
    assert torch.equal(e0, e0b)  # deterministic per (seed, epoch)
