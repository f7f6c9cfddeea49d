import numpy as np

from cilfw.data import (ClassIncremental, make_synthetic, CIFAR100_CLASS_ORDER,
                        build_dataset)
from cilfw.config import parse_args


def test_class_order_is_permutation():
    assert sorted(CIFAR100_CLASS_ORDER) == list(range(100))


def _scenario(nc=20, per_class=10, base=10, inc=5, order=None):
    x, y = make_synthetic(nc, per_class, 8, seed=0)
    return ClassIncremental(x, y, base, inc, class_order=order)


def test_task_split_counts():
    s = _scenario()
    assert len(s) == 3
    assert s.increments(0) == 10
    assert s.increments(1) == 5
    assert len(s[0]) == 100
    assert len(s[1]) == 50


def test_label_remap_follows_class_order():
    order = list(range(19, -1, -1))  # reversed
    s = _scenario(order=order)
    t0 = s[0]
    # task 0 = first 10 entries of class_order -> remapped labels 0..9
    assert set(np.unique(t0.y)) == set(range(10))
    # original class 19 should be remapped label 0
    x, y = make_synthetic(20, 10, 8, seed=0)
    mask19 = y == 19
    assert mask19.sum() == 10
    assert (np.sort(np.unique(t0.y)) == np.arange(10)).all()


def test_slice_merges_tasks():
    s = _scenario()
    merged = s[:2]
    assert len(merged) == 150
    assert set(np.unique(merged.y)) == set(range(15))
    assert set(np.unique(merged.t)) == {0, 1}


def test_add_samples_and_raw():
    s = _scenario()
    t1 = s[1]
    n0 = len(t1)
    x, y, t = s[0].get_raw_samples()
    t1.add_samples(x[:7], y[:7], t[:7])
    assert len(t1) == n0 + 7
    rx, ry, rt = t1.get_raw_samples()
    assert rx.shape[0] == n0 + 7
    assert rx.dtype == np.uint8


def test_b0_protocol():
    # num_bases=0 -> initial task uses `increment` classes (B0-Inc10)
    x, y = make_synthetic(100, 4, 8, seed=0)
    s = ClassIncremental(x, y, 0, 10)
    assert len(s) == 10
    assert s.increments(0) == 10


def test_build_dataset_synthetic():
    args = parse_args(["--data_set", "synthetic", "--num_bases", "50",
                       "--increment", "10", "--synthetic_train_size", "400",
                       "--input_size", "8"])
    scenario, nb = build_dataset(True, args)
    assert nb == 100
    assert len(scenario) == 6
    img, label, tid = scenario[0][0]
    assert img.shape == (8, 8, 3)
    assert tid == 0
