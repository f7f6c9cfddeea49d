import numpy as np
import torch

from cilfw.config import parse_args
from cilfw.data.transforms import (TrainTransform, EvalTransform, RandAugment,
                                   RAND_AUGMENT_OPS, random_crop_pad,
                                   RandomErasing)


def _img(size=32, seed=0):
    rng = np.random.default_rng(seed)
    return rng.integers(0, 255, size=(size, size, 3), dtype=np.uint8)


def test_train_transform_shape_and_dtype():
    args = parse_args(["--input_size", "32"])
    tf = TrainTransform(args, "cifar100")
    out = tf(_img())
    assert out.shape == (32, 32, 3)
    assert out.dtype == torch.float32
    assert torch.isfinite(out).all()


def test_eval_transform_normalizes():
    args = parse_args(["--input_size", "32"])
    tf = EvalTransform(args, "cifar100")
    out = tf(_img())
    assert out.shape == (32, 32, 3)
    # normalized data should be roughly centered
    assert out.abs().mean() < 3.0


def test_eval_transform_center_crop_large():
    args = parse_args(["--input_size", "224"])
    tf = EvalTransform(args, "imagenet")
    out = tf(_img(256))
    assert out.shape == (224, 224, 3)


def test_all_randaugment_ops_valid():
    img = _img()
    for op in RAND_AUGMENT_OPS:
        np.random.seed(0)
        out = op(img.copy(), 9.0)
        assert out.shape == img.shape
        assert out.dtype == np.uint8


def test_randaugment_policy_parse():
    ra = RandAugment.from_policy("rand-m9-mstd0.5-inc1")
    assert ra.magnitude == 9 and ra.mstd == 0.5 and ra.num_ops == 2
    np.random.seed(1)
    out = ra(_img())
    assert out.shape == (32, 32, 3)


def test_random_crop_pad():
    np.random.seed(0)
    out = random_crop_pad(_img(), 32, padding=4)
    assert out.shape == (32, 32, 3)


def test_random_erasing_changes_pixels():
    t = torch.zeros(32, 32, 3)
    er = RandomErasing(prob=1.0, mode="pixel")
    np.random.seed(0)
    torch.manual_seed(0)
    out = er(t.clone())
    assert (out != 0).any()


def test_no_aug_flag():
    args = parse_args(["--input_size", "32", "--no_aug"])
    tf = TrainTransform(args, "cifar100")
    img = _img()
    o1, o2 = tf(img), tf(img)
    assert torch.equal(o1, o2)
