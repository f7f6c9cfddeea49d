import os

import torch

from cilfw.models import CilModel
from cilfw.cil import (RehearsalMemory, save_task_checkpoint,
                       load_task_checkpoint)
import numpy as np


class _Args:
    task_id = 1
    known_classes = 5
    nb_classes = 10
    class_order = list(range(10))


def test_checkpoint_roundtrip(tmp_path):
    torch.manual_seed(0)
    model = CilModel("resnet20", 32)
    model.prev_model_adaption(5)
    model.prev_model_adaption(5)
    mem = RehearsalMemory(memory_size=8)
    x = np.random.randint(0, 255, (20, 4, 4, 3), dtype=np.uint8)
    y = np.repeat([0, 1], 10)
    t = np.zeros(20, dtype=np.int64)
    mem.add(x, y, t, torch.randn(20, 4))
    args = _Args()

    path = save_task_checkpoint(str(tmp_path), 1, model, mem, [50.0, 40.0],
                                args)
    assert os.path.exists(path)

    model2 = CilModel("resnet20", 32)
    mem2 = RehearsalMemory()
    args2 = _Args()
    args2.task_id = args2.known_classes = 0
    state = load_task_checkpoint(path, model2, mem2, args2)

    assert state["task_id"] == 1
    assert args2.known_classes == 5
    assert [h.out_features for h in model2.fc.heads] == [5, 5]
    for (k1, v1), (k2, v2) in zip(model.state_dict().items(),
                                  model2.state_dict().items()):
        assert k1 == k2 and torch.equal(v1, v2.to(v1.dtype)), k1
    assert len(mem2) == len(mem)
    mx1, my1, _ = mem.get()
    mx2, my2, _ = mem2.get()
    assert (mx1 == mx2).all() and (my1 == my2).all()
    assert state["acc1s"] == [50.0, 40.0]


def test_checkpoint_restores_rng(tmp_path):
    model = CilModel("resnet20", 32)
    model.prev_model_adaption(3)
    args = _Args()
    torch.manual_seed(7)
    _ = torch.randn(3)  # advance
    mem = RehearsalMemory(memory_size=4)
    mem.add(np.zeros((4, 2, 2, 3), dtype=np.uint8),
            np.array([0, 0, 1, 1]), np.zeros(4, dtype=np.int64),
            torch.randn(4, 2))
    path = save_task_checkpoint(str(tmp_path), 0, model, mem, [1.0], args)
    expected_next = torch.randn(2)  # what the RNG yields after saving

    torch.manual_seed(999)  # scramble
    model2 = CilModel("resnet20", 32)
    mem2 = RehearsalMemory()
    args2 = _Args()
    load_task_checkpoint(path, model2, mem2, args2, restore_rng=True)
    got = torch.randn(2)
    assert torch.equal(got, expected_next)
