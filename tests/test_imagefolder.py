"""Lazy ImageFolder source: scan -> scenario split -> decode at __getitem__ ->
rehearsal memory stores paths -> DeviceExemplarStore decodes once."""

import os

import numpy as np
import pytest
import torch
from PIL import Image

from cilfw.data.datasets import scan_imagefolder, load_image
from cilfw.data.scenario import ClassIncremental
from cilfw.cil import RehearsalMemory
from cilfw.cil.replay_gpu import DeviceExemplarStore


@pytest.fixture()
def folder(tmp_path):
    rng = np.random.default_rng(0)
    for ci in range(4):
        d = tmp_path / "train" / f"class_{ci}"
        d.mkdir(parents=True)
        for j in range(6):
            arr = rng.integers(0, 255, (12, 12, 3), dtype=np.uint8)
            Image.fromarray(arr).save(d / f"img_{j}.png")
    return str(tmp_path)


def test_scan_and_decode(folder):
    paths, labels, nc = scan_imagefolder(folder, "train")
    assert nc == 4 and len(paths) == 24
    img = load_image(paths[0])
    assert img.shape == (12, 12, 3) and img.dtype == np.uint8


def test_paths_flow_through_scenario_and_memory(folder):
    paths, labels, nc = scan_imagefolder(folder, "train")
    scenario = ClassIncremental(paths, labels, 2, 1)
    assert len(scenario) == 3
    t0 = scenario[0]
    img, y, tid = t0[0]
    assert img.shape[0] == 12 and tid == 0  # decoded lazily

    mem = RehearsalMemory(memory_size=8)
    x, yy, tt = t0.get_raw_samples()
    mem.add(x, yy, tt, torch.randn(len(yy), 8))
    mx, my, _ = mem.get()
    assert mx.dtype == object  # memory stores PATHS, not pixels

    # HBM store decodes the paths once
    store = DeviceExemplarStore.from_memory(mem, "synthetic", device="cpu")
    imgs, lab = store.sample(4, augment=False, dtype=torch.float32)
    assert imgs.shape == (4, 12, 12, 3)
