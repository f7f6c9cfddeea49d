"""Deterministic distributed sharding — torch DistributedSampler semantics.

Reference uses torch's DistributedSampler (template.py:232-235) incl. its
pad-by-repetition on uneven shards (SURVEY.md §3.2 notes eval-accuracy depends on
this). Reimplemented first-party so the sharding contract is explicit and testable
without torch internals.

``compute_shard`` is the single implementation of the contract; both the
host-path ``DistributedSampler`` and the device-path ``GpuTaskLoader``
(cilfw/data/gpu_pipeline.py) call it, so the two loader paths cannot drift.
"""

import math

import torch


def compute_shard(n, num_replicas, rank, shuffle=True, seed=0, epoch=0,
                  drop_last=False):
    """This rank's sample indices as an int64 CPU tensor.

    Exact torch.utils.data.DistributedSampler semantics: a seed+epoch
    torch.randperm over all n samples, padded by repetition from the front
    (or truncated under drop_last) to a multiple of num_replicas, then the
    rank-strided slice indices[rank::num_replicas].
    """
    if shuffle:
        g = torch.Generator()
        g.manual_seed(seed + epoch)
        idx = torch.randperm(n, generator=g)
    else:
        idx = torch.arange(n)
    if drop_last and n % num_replicas != 0:
        num_samples = n // num_replicas
    else:
        num_samples = math.ceil(n / num_replicas)
    total = num_samples * num_replicas
    if total > n:  # pad by repetition from the front (torch semantics)
        reps = (total - n + n - 1) // n
        idx = torch.cat([idx] + [idx] * reps)[:total]
    else:
        idx = idx[:total]
    return idx[rank:total:num_replicas]


class DistributedSampler(torch.utils.data.Sampler):
    def __init__(self, dataset, num_replicas, rank, shuffle=True, seed=0,
                 drop_last=False):
        self.dataset = dataset
        self.num_replicas = num_replicas
        self.rank = rank
        self.shuffle = shuffle
        self.seed = seed
        self.drop_last = drop_last
        self.epoch = 0
        n = len(dataset)
        if drop_last and n % num_replicas != 0:
            self.num_samples = n // num_replicas
        else:
            self.num_samples = math.ceil(n / num_replicas)
        self.total_size = self.num_samples * num_replicas

    def set_epoch(self, epoch):
        self.epoch = epoch

    def __len__(self):
        return self.num_samples

    def __iter__(self):
        shard = compute_shard(len(self.dataset), self.num_replicas, self.rank,
                              self.shuffle, self.seed, self.epoch,
                              self.drop_last)
        assert shard.numel() == self.num_samples
        return iter(shard.tolist())
