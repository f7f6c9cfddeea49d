"""HBM-resident exemplar store — the 288 GB residency component (BASELINE.md
config 4: ImageNet-1000 exemplars + frozen teacher resident in HBM).

The reference round-trips replay exemplars through CPU numpy + the DataLoader
every epoch (template.py:230-231, 292-299). On MI355X, 288 GB HBM3E holds the
entire exemplar set (20k ImageNet exemplars ≈ 3 GB raw; even the full 1.28M
train images at 224² ≈ 193 GB) — so cilfw keeps the decoded exemplar tensors
on-device and assembles replay batches with zero host traffic:

    store = DeviceExemplarStore.from_memory(memory, args, device="cuda")
    imgs, labels = store.sample(n)   # bf16 NHWC normalized, on-device

Augmentation on sampled batches is the device-side subset (random crop pad +
horizontal flip via torch index ops); the full RandAugment pipeline stays on
the host path (SURVEY.md §2.3 K14 — not kernel-critical).
"""

import numpy as np
import torch

from ..data.datasets import DATASET_STATS, load_image


class DeviceExemplarStore:
    def __init__(self, images_u8, labels, mean, std, device="cuda",
                 generator=None):
        """images_u8: uint8 (N,H,W,C) tensor/array; labels: int64 (N,)."""
        if isinstance(images_u8, np.ndarray):
            images_u8 = torch.from_numpy(np.ascontiguousarray(images_u8))
        self.images = images_u8.to(device)
        self.labels = torch.as_tensor(labels, dtype=torch.int64).to(device)
        self.mean = torch.tensor(mean, device=device).view(1, 1, 1, -1) * 255.0
        self.std = torch.tensor(std, device=device).view(1, 1, 1, -1) * 255.0
        self.device = device
        self.generator = generator

    @classmethod
    def from_memory(cls, memory, stats_key="cifar100", device="cuda",
                    image_size=None):
        """Upload a RehearsalMemory's exemplars (decoding path-based stores)."""
        x, y, _t = memory.get()
        if x.dtype == object:  # lazy path store -> decode once, stay resident
            imgs = np.stack([load_image(p) for p in x])
        else:
            imgs = x
        if image_size is not None and imgs.shape[1] != image_size:
            t = torch.from_numpy(imgs).permute(0, 3, 1, 2).float()
            t = torch.nn.functional.interpolate(t, size=(image_size,
                                                         image_size),
                                                mode="bilinear")
            imgs = t.permute(0, 2, 3, 1).to(torch.uint8).numpy()
        mean, std = DATASET_STATS[stats_key]
        return cls(imgs, y, mean, std, device=device)

    def __len__(self):
        return self.images.shape[0]

    @property
    def nbytes(self):
        return self.images.numel() * self.images.element_size()

    def sample(self, n, augment=True, dtype=torch.bfloat16, pad=4):
        """Replay batch fully on-device: gather -> (crop+flip) -> normalize."""
        idx = torch.randint(0, len(self), (n,), device=self.device,
                            generator=self.generator)
        imgs = self.images[idx].float()
        if augment:
            N, H, W, C = imgs.shape
            # random crop with zero padding
            padded = torch.zeros(N, H + 2 * pad, W + 2 * pad, C,
                                 device=self.device)
            padded[:, pad:pad + H, pad:pad + W] = imgs
            oy = torch.randint(0, 2 * pad + 1, (N,), device=self.device,
                               generator=self.generator)
            ox = torch.randint(0, 2 * pad + 1, (N,), device=self.device,
                               generator=self.generator)
            rows = oy.view(N, 1) + torch.arange(H, device=self.device)
            cols = ox.view(N, 1) + torch.arange(W, device=self.device)
            imgs = padded[torch.arange(N, device=self.device).view(N, 1, 1),
                          rows.view(N, H, 1), cols.view(N, 1, W)]
            # horizontal flip half the batch
            flip = torch.rand(N, device=self.device,
                              generator=self.generator) < 0.5
            imgs[flip] = imgs[flip].flip(2)
        imgs = (imgs - self.mean) / self.std
        return imgs.to(dtype), self.labels[idx]


class DeviceReplayMirror:
    """Device-resident mirror of a RehearsalMemory's exemplar *images* —
    the engine's default replay source under ``--gpu_data``.

    The reference re-materializes replay on the host every task
    (memory.get() -> numpy concat -> DataLoader, template.py:230-231); here
    exemplars selected from an already-uploaded task tensor are gathered ON
    DEVICE (``update`` with ``task_images``) and stay resident in HBM across
    tasks, so the replay fraction of every batch never touches the host.
    Content is kept exactly equal to the host RehearsalMemory (same herding
    keep-indices via ``memory.last_selection``, same quota trimming, same
    sorted-class concat order as ``memory.get()``), which the engine asserts.
    """

    def __init__(self, device):
        self.device = device
        self._imgs = {}    # class -> uint8 (k,H,W,C) device tensor, ranked
        self._labels = {}  # class -> int64 (k,) device tensor
        self._tasks = {}   # class -> int64 (k,) device tensor

    def update(self, memory, task_images=None, task_id=0):
        """Ingest memory's last add(): gather new-class exemplars from the
        device-resident ``task_images`` (uint8, aligned with the x passed to
        memory.add), then trim every class to the new quota."""
        for c, keep in memory.last_selection.items():
            if task_images is not None:
                idx = torch.as_tensor(np.ascontiguousarray(keep),
                                      dtype=torch.int64,
                                      device=task_images.device)
                self._imgs[c] = task_images[idx].to(self.device)
            else:  # fallback: one H2D upload of the selected exemplars
                self._imgs[c] = torch.from_numpy(
                    np.ascontiguousarray(memory._x[c])).to(self.device)
            self._labels[c] = torch.from_numpy(
                np.ascontiguousarray(memory._y[c]).astype(np.int64)
            ).to(self.device)
            self._tasks[c] = torch.from_numpy(
                np.ascontiguousarray(memory._t[c]).astype(np.int64)
            ).to(self.device)
        q = memory.last_quota
        for c in list(self._imgs):
            self._imgs[c] = self._imgs[c][:q]
            self._labels[c] = self._labels[c][:q]
            self._tasks[c] = self._tasks[c][:q]

    @classmethod
    def from_memory(cls, memory, device):
        """Rebuild after checkpoint resume (one upload, then resident)."""
        self = cls(device)
        for c in sorted(memory._x):
            self._imgs[c] = torch.from_numpy(
                np.ascontiguousarray(memory._x[c])).to(device)
            self._labels[c] = torch.from_numpy(
                np.ascontiguousarray(memory._y[c]).astype(np.int64)).to(device)
            self._tasks[c] = torch.from_numpy(
                np.ascontiguousarray(memory._t[c]).astype(np.int64)).to(device)
        return self

    def __len__(self):
        return sum(v.shape[0] for v in self._labels.values())

    def get(self):
        """(images_u8, labels, task_ids) device tensors, memory.get() order."""
        classes = sorted(self._imgs)
        if not classes:
            raise ValueError("mirror is empty")
        return (torch.cat([self._imgs[c] for c in classes]),
                torch.cat([self._labels[c] for c in classes]),
                torch.cat([self._tasks[c] for c in classes]))
