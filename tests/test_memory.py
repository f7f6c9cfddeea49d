import numpy as np
import torch

from cilfw.cil import RehearsalMemory, herding_select


def naive_herding(f, m):
    """Literal iCaRL greedy loop (the reference behavior)."""
    f = f.double().numpy()
    mu = f.mean(axis=0)
    sel, sum_sel = [], np.zeros_like(mu)
    for k in range(m):
        best, bi = None, None
        for i in range(len(f)):
            if i in sel:
                continue
            d = np.linalg.norm(mu - (sum_sel + f[i]) / (k + 1))
            if best is None or d < best - 1e-12:
                best, bi = d, i
        sel.append(bi)
        sum_sel += f[bi]
    return sel


def test_herding_matches_naive():
    torch.manual_seed(0)
    f = torch.randn(40, 8)
    got = herding_select(f, 10).tolist()
    want = naive_herding(f, 10)
    assert got == want


def test_herding_first_pick_is_closest_to_mean():
    torch.manual_seed(1)
    f = torch.randn(30, 4)
    mu = f.mean(0)
    d = (f - mu).norm(dim=1)
    assert herding_select(f, 1).item() == d.argmin().item()


def _fake_task(classes, per_class=20, seed=0):
    rng = np.random.default_rng(seed)
    x = rng.integers(0, 255, size=(per_class * len(classes), 4, 4, 3),
                     dtype=np.uint8)
    y = np.repeat(classes, per_class)
    t = np.zeros_like(y)
    feats = torch.randn(len(y), 8)
    return x, y, t, feats


def test_memory_quota_shrinks():
    mem = RehearsalMemory(memory_size=40, herding_method="barycenter")
    x, y, t, f = _fake_task([0, 1])
    mem.add(x, y, t, f)
    assert len(mem) == 40  # 20 per class
    x, y, t, f = _fake_task([2, 3], seed=1)
    mem.add(x, y, t, f)
    assert len(mem) == 40  # 10 per class now
    mx, my, mt = mem.get()
    counts = np.bincount(my)
    assert (counts[:4] == 10).all()


def test_memory_fixed_quota():
    mem = RehearsalMemory(memory_size=40, fixed_memory=True,
                          nb_total_classes=10)
    x, y, t, f = _fake_task([0, 1])
    mem.add(x, y, t, f)
    assert len(mem) == 8  # 4 per class fixed
    x, y, t, f = _fake_task([2], seed=2)
    mem.add(x, y, t, f)
    counts = np.bincount(mem.get()[1])
    assert (counts[:3] == 4).all()


def test_memory_keeps_herding_rank_prefix():
    """Shrinking must keep the TOP-ranked exemplars (herding order)."""
    mem = RehearsalMemory(memory_size=20)
    x, y, t, f = _fake_task([0])
    mem.add(x, y, t, f)
    kept_20 = mem._x[0].copy()
    mem2 = RehearsalMemory(memory_size=40)
    mem2.add(x, y, t, f)
    # first 20 of the 40-budget selection == the 20-budget selection
    assert (mem2._x[0][:20] == kept_20).all()


def test_memory_ignores_already_stored_classes():
    mem = RehearsalMemory(memory_size=40)
    x, y, t, f = _fake_task([0, 1])
    mem.add(x, y, t, f)
    stored = mem._x[0].copy()
    # re-adding the same classes (e.g. replayed samples in task data) is a no-op
    mem.add(x, y, t, f)
    assert (mem._x[0][:len(stored)] == stored[:len(mem._x[0])]).all()


def test_random_herding_deterministic():
    m1 = RehearsalMemory(memory_size=10, herding_method="random")
    m2 = RehearsalMemory(memory_size=10, herding_method="random")
    x, y, t, f = _fake_task([0])
    m1.add(x, y, t, f)
    m2.add(x, y, t, f)
    assert (m1.get()[0] == m2.get()[0]).all()
