import torch

from cilfw.data.sampler import DistributedSampler


class _DS(torch.utils.data.Dataset):
    def __init__(self, n):
        self.n = n

    def __len__(self):
        return self.n

    def __getitem__(self, i):
        return i


def test_matches_torch_distributed_sampler():
    ds = _DS(103)
    for world in (1, 2, 4):
        for rank in range(world):
            ours = DistributedSampler(ds, world, rank, shuffle=True, seed=7)
            ref = torch.utils.data.DistributedSampler(
                ds, num_replicas=world, rank=rank, shuffle=True, seed=7)
            ours.set_epoch(3)
            ref.set_epoch(3)
            assert list(ours) == list(ref)


def test_padding_covers_all_and_equal_shards():
    ds = _DS(10)
    world = 4
    shards = [list(DistributedSampler(ds, world, r, shuffle=False))
              for r in range(world)]
    assert all(len(s) == 3 for s in shards)
    assert set(i for s in shards for i in s) == set(range(10))


def test_drop_last():
    ds = _DS(10)
    shards = [list(DistributedSampler(ds, 4, r, shuffle=False, drop_last=True))
              for r in range(4)]
    assert all(len(s) == 2 for s in shards)
