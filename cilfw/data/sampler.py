"""Deterministic distributed sampler — torch DistributedSampler semantics.

Reference uses torch's DistributedSampler (template.py:232-235) incl. its
pad-by-repetition on uneven shards (SURVEY.md §3.2 notes eval-accuracy depends on
this). Reimplemented first-party so the sharding contract is explicit and testable
without torch internals.
"""

import math

import torch


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
        n = len(self.dataset)
        if self.shuffle:
            g = torch.Generator()
            g.manual_seed(self.seed + self.epoch)
            indices = torch.randperm(n, generator=g).tolist()
        else:
            indices = list(range(n))
        if not self.drop_last:
            pad = self.total_size - len(indices)
            if pad > 0:
                # torch semantics: repeat from the front
                reps = math.ceil(pad / n)
                indices += (indices * reps)[:pad]
        else:
            indices = indices[:self.total_size]
        assert len(indices) == self.total_size
        shard = indices[self.rank:self.total_size:self.num_replicas]
        assert len(shard) == self.num_samples
        return iter(shard)
