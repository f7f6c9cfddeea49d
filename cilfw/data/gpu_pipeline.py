"""Device-resident data pipeline — feeds the HIP training step at full rate.

A Python-side DataLoader tops out far below the ~20k imgs/s the MI355X step
consumes at CIFAR scale. When the task's images are array-backed (CIFAR /
synthetic; not lazy ImageFolder paths), cilfw uploads the task set ONCE as
uint8 (CIFAR-100 full train split = 150 MB — nothing against 288 GB HBM) and
assembles every batch on-device: gather -> random crop(+pad) -> horizontal
flip -> [RandAugment / color-jitter / random-erasing, see device_augment.py]
-> normalize -> bf16, all torch index ops on the GPU.

Sharding is the torch-DistributedSampler contract via the single shared
implementation ``cilfw.data.sampler.compute_shard`` (same seed+epoch
permutation on every rank, pad-by-repetition, rank-strided slice).

``extra`` accepts already-device-resident (images_u8, labels, task_ids)
tensors — the DeviceReplayMirror replay path (cilfw/cil/replay_gpu.py):
exemplars are concatenated after the task tensor exactly like the
reference's TaskSet.add_samples appends them (template.py:230-231), so the
epoch shuffle sees an identically-ordered index space.
"""

import numpy as np
import torch

from .sampler import compute_shard


class GpuTaskLoader:
    def __init__(self, taskset, batch_size, device, mean, std, world=1, rank=0,
                 shuffle=True, seed=0, augment=True, drop_last=True,
                 dtype=torch.bfloat16, pad=4, extra=None, aug_pipeline=None):
        assert taskset.x.dtype == np.uint8, \
            "GpuTaskLoader needs array-backed images (not lazy paths)"
        images = torch.from_numpy(np.ascontiguousarray(taskset.x)).to(device)
        labels = torch.from_numpy(
            np.ascontiguousarray(taskset.y).astype(np.int64)).to(device)
        self.n_task = images.shape[0]
        if extra is not None:
            ex_imgs, ex_labels = extra[0], extra[1]
            assert ex_imgs.dtype == torch.uint8
            images = torch.cat([images, ex_imgs.to(device)])
            labels = torch.cat([labels, ex_labels.to(device,
                                                     torch.int64)])
        self.images = images
        self.labels = labels
        self.batch_size = batch_size
        self.device = device
        self.world, self.rank = world, rank
        self.shuffle, self.seed = shuffle, seed
        self.augment = augment
        self.aug_pipeline = aug_pipeline  # optional DeviceAugment (RandAugment etc.)
        self.drop_last = drop_last
        self.dtype = dtype
        self.pad = pad
        self.epoch = 0
        self.mean = (torch.tensor(mean, device=device).view(1, 1, 1, -1)
                     * 255.0)
        self.std = (torch.tensor(std, device=device).view(1, 1, 1, -1)
                    * 255.0)
        n = len(self.labels)
        import math
        if drop_last and n % world != 0:
            self.num_samples = n // world
        else:
            self.num_samples = math.ceil(n / world)

    def set_epoch(self, epoch):
        self.epoch = epoch

    def __len__(self):
        if self.drop_last:
            return self.num_samples // self.batch_size
        return (self.num_samples + self.batch_size - 1) // self.batch_size

    def _shard_indices(self):
        shard = compute_shard(len(self.labels), self.world, self.rank,
                              self.shuffle, self.seed, self.epoch,
                              self.drop_last)
        return shard.to(self.device)

    def __iter__(self):
        shard = self._shard_indices()
        g = torch.Generator(device=self.device)
        g.manual_seed(self.seed * 1000003 + self.epoch * 131 + self.rank)
        nb = len(self)
        aug = self.aug_pipeline if self.augment else None
        aug_imgs = None
        if aug is not None and (aug.enabled or aug.color_jitter > 0):
            # EPOCH-level batched RandAugment + jitter over the whole shard:
            # per-batch application is host-sync/launch-bound (~200 launches
            # + op-subset syncs per 128-batch cost 2/3 of step throughput);
            # one pass per epoch amortizes that ~40x. Results are integral
            # 0..255 so they round-trip through uint8 exactly. (Order
            # deviation vs the host pipeline: RA/jitter run before the
            # per-batch pad-crop/flip instead of after — documented.)
            srcs = self.images[shard]
            outs = []
            for c0 in range(0, srcs.shape[0], 4096):
                outs.append(aug(srcs[c0:c0 + 4096], g).to(torch.uint8))
            aug_imgs = torch.cat(outs)
            del outs, srcs
        for b in range(nb):
            idx = shard[b * self.batch_size:(b + 1) * self.batch_size]
            if aug_imgs is not None:
                imgs = aug_imgs[b * self.batch_size:
                                (b + 1) * self.batch_size].float()
            else:
                imgs = self.images[idx].float()
            if self.augment:
                imgs = self._crop_flip(imgs, g)
            imgs = ((imgs - self.mean) / self.std).to(self.dtype)
            if aug is not None and aug.reprob > 0:
                imgs = aug.erase(imgs, g)
            yield imgs, self.labels[idx], None

    def _crop_flip(self, imgs, g):
        N, H, W, C = imgs.shape
        p = self.pad
        padded = torch.zeros(N, H + 2 * p, W + 2 * p, C, device=self.device)
        padded[:, p:p + H, p:p + W] = imgs
        oy = torch.randint(0, 2 * p + 1, (N,), device=self.device,
                           generator=g)
        ox = torch.randint(0, 2 * p + 1, (N,), device=self.device,
                           generator=g)
        rows = oy.view(N, 1) + torch.arange(H, device=self.device)
        cols = ox.view(N, 1) + torch.arange(W, device=self.device)
        imgs = padded[torch.arange(N, device=self.device).view(N, 1, 1),
                      rows.view(N, H, 1), cols.view(N, 1, W)]
        flip = torch.rand(N, device=self.device, generator=g) < 0.5
        imgs[flip] = imgs[flip].flip(2)
        return imgs
