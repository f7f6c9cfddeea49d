"""Augmentation pipeline — rebuilds the reference's timm transform stack from scratch.

Reference contract (utils.py:210-251 via timm.data.create_transform): train =
RandomCrop(pad 4, small inputs) / RandomResizedCrop (large), horizontal flip,
color-jitter 0.4, RandAugment policy ``rand-m9-mstd0.5-inc1`` (2 ops/image,
magnitude 9, magnitude-std 0.5, increasing-severity arg schedule), normalize,
RandomErasing(prob/mode/count). Eval = resize+center-crop (large) / identity
(small), normalize.

Implemented on uint8 HWC numpy arrays (scipy.ndimage for the affine ops); output is
a float32 HWC torch tensor (cilfw is NHWC end-to-end).
"""

import numpy as np
import torch

try:
    from scipy import ndimage as ndi
except ImportError:  # pragma: no cover
    ndi = None

from .datasets import DATASET_STATS


# ------------------------------------------------------------------ RandAugment ops

def _blend(a, b, factor):
    out = b.astype(np.float32) + factor * (a.astype(np.float32) - b.astype(np.float32))
    return np.clip(out, 0, 255).astype(np.uint8)


def _autocontrast(img, _):
    out = img.astype(np.float32).copy()
    for c in range(img.shape[2]):
        lo, hi = out[..., c].min(), out[..., c].max()
        if hi > lo:
            out[..., c] = (out[..., c] - lo) * (255.0 / (hi - lo))
    return out.astype(np.uint8)


def _equalize(img, _):
    out = img.copy()
    for c in range(img.shape[2]):
        ch = out[..., c]
        hist = np.bincount(ch.ravel(), minlength=256)
        nonzero = hist[hist > 0]
        if len(nonzero) <= 1:
            continue
        step = (hist.sum() - nonzero[-1]) // 255
        if step == 0:
            continue
        lut = (np.cumsum(hist) - hist) // step
        out[..., c] = np.clip(lut, 0, 255).astype(np.uint8)[ch]
    return out


def _invert(img, _):
    return 255 - img


def _rotate(img, mag):
    deg = mag / 10.0 * 30.0 * (1 if np.random.rand() < 0.5 else -1)
    if ndi is None:
        return img
    return np.clip(ndi.rotate(img.astype(np.float32), deg, axes=(0, 1),
                              reshape=False, order=1, mode="constant", cval=128),
                   0, 255).astype(np.uint8)


def _posterize(img, mag):
    bits = max(1, 8 - int(round(mag / 10.0 * 4)))
    mask = ~np.uint8((1 << (8 - bits)) - 1)
    return img & mask


def _solarize(img, mag):
    thr = 256 - int(round(mag / 10.0 * 256))
    return np.where(img >= thr, 255 - img, img).astype(np.uint8)


def _solarize_add(img, mag):
    add = int(round(mag / 10.0 * 110))
    out = img.astype(np.int32)
    out = np.where(out < 128, np.clip(out + add, 0, 255), out)
    return out.astype(np.uint8)


def _enhance_factor(mag):
    f = mag / 10.0 * 0.9
    return 1.0 + (f if np.random.rand() < 0.5 else -f)


def _color(img, mag):
    gray = img.mean(axis=2, keepdims=True).repeat(img.shape[2], axis=2)
    return _blend(img, gray, _enhance_factor(mag))


def _contrast(img, mag):
    mean = np.full_like(img, int(img.astype(np.float32).mean()))
    return _blend(img, mean, _enhance_factor(mag))


def _brightness(img, mag):
    return _blend(img, np.zeros_like(img), _enhance_factor(mag))


def _sharpness(img, mag):
    if ndi is None:
        return img
    kernel = np.array([[1, 1, 1], [1, 5, 1], [1, 1, 1]], dtype=np.float32) / 13.0
    smooth = np.stack([ndi.convolve(img[..., c].astype(np.float32), kernel,
                                    mode="nearest")
                       for c in range(img.shape[2])], axis=2)
    return _blend(img, smooth.astype(np.uint8), _enhance_factor(mag))


def _affine(img, matrix, offset):
    if ndi is None:
        return img
    out = np.stack([ndi.affine_transform(img[..., c].astype(np.float32), matrix,
                                         offset=offset, order=1, mode="constant",
                                         cval=128)
                    for c in range(img.shape[2])], axis=2)
    return np.clip(out, 0, 255).astype(np.uint8)


def _shear_x(img, mag):
    s = mag / 10.0 * 0.3 * (1 if np.random.rand() < 0.5 else -1)
    return _affine(img, np.array([[1, s], [0, 1]]), (-s * img.shape[1] / 2, 0))


def _shear_y(img, mag):
    s = mag / 10.0 * 0.3 * (1 if np.random.rand() < 0.5 else -1)
    return _affine(img, np.array([[1, 0], [s, 1]]), (0, -s * img.shape[0] / 2))


def _translate_x(img, mag):
    t = mag / 10.0 * 0.45 * img.shape[1] * (1 if np.random.rand() < 0.5 else -1)
    return _affine(img, np.eye(2), (0, t))


def _translate_y(img, mag):
    t = mag / 10.0 * 0.45 * img.shape[0] * (1 if np.random.rand() < 0.5 else -1)
    return _affine(img, np.eye(2), (t, 0))


RAND_AUGMENT_OPS = [
    _autocontrast, _equalize, _invert, _rotate, _posterize, _solarize,
    _solarize_add, _color, _contrast, _brightness, _sharpness, _shear_x, _shear_y,
    _translate_x, _translate_y,
]


class RandAugment:
    """Policy 'rand-mM-mstdS-inc1': N random ops at magnitude ~N(M, S*?).

    timm semantics: magnitude jittered per-op by N(0, mstd) when mstd>0."""

    def __init__(self, num_ops=2, magnitude=9, mstd=0.5):
        self.num_ops, self.magnitude, self.mstd = num_ops, magnitude, mstd

    def __call__(self, img):
        for _ in range(self.num_ops):
            op = RAND_AUGMENT_OPS[np.random.randint(len(RAND_AUGMENT_OPS))]
            mag = self.magnitude
            if self.mstd > 0:
                mag = float(np.clip(np.random.normal(mag, self.mstd), 0, 10))
            img = op(img, mag)
        return img

    @classmethod
    def from_policy(cls, policy):
        """Parse 'rand-m9-mstd0.5-inc1' (the reference default, template.py:25)."""
        mag, mstd, n = 9, 0.5, 2
        for part in policy.split("-"):
            if part.startswith("mstd"):
                mstd = float(part[4:])
            elif part.startswith("m") and part[1:].replace(".", "").isdigit():
                mag = float(part[1:])
            elif part.startswith("n") and part[1:].isdigit():
                n = int(part[1:])
        return cls(num_ops=n, magnitude=mag, mstd=mstd)


# --------------------------------------------------------------------- base crops

def random_crop_pad(img, size, padding=4):
    padded = np.pad(img, ((padding, padding), (padding, padding), (0, 0)),
                    mode="constant")
    y = np.random.randint(0, padded.shape[0] - size + 1)
    x = np.random.randint(0, padded.shape[1] - size + 1)
    return padded[y:y + size, x:x + size]


def random_resized_crop(img, size, scale=(0.08, 1.0), ratio=(3 / 4, 4 / 3)):
    H, W = img.shape[:2]
    for _ in range(10):
        area = H * W * np.random.uniform(*scale)
        ar = np.exp(np.random.uniform(np.log(ratio[0]), np.log(ratio[1])))
        w = int(round(np.sqrt(area * ar)))
        h = int(round(np.sqrt(area / ar)))
        if w <= W and h <= H:
            y = np.random.randint(0, H - h + 1)
            x = np.random.randint(0, W - w + 1)
            return _resize(img[y:y + h, x:x + w], size)
    return _resize(_center_crop(img, min(H, W)), size)


def _resize(img, size):
    if img.shape[0] == size and img.shape[1] == size:
        return img
    if ndi is None:
        idx_y = np.linspace(0, img.shape[0] - 1, size).astype(int)
        idx_x = np.linspace(0, img.shape[1] - 1, size).astype(int)
        return img[idx_y][:, idx_x]
    zoom = (size / img.shape[0], size / img.shape[1], 1)
    return np.clip(ndi.zoom(img.astype(np.float32), zoom, order=1), 0,
                   255).astype(np.uint8)


def _center_crop(img, size):
    H, W = img.shape[:2]
    y, x = (H - size) // 2, (W - size) // 2
    return img[y:y + size, x:x + size]


def color_jitter(img, strength):
    if strength <= 0:
        return img
    for fn in (_brightness, _contrast, _color):
        f = 1.0 + np.random.uniform(-strength, strength)
        base = {_brightness: np.zeros_like(img),
                _contrast: np.full_like(img, int(img.astype(np.float32).mean())),
                _color: img.mean(axis=2, keepdims=True).repeat(img.shape[2],
                                                               axis=2)}[fn]
        img = _blend(img, base, f)
    return img


class RandomErasing:
    """'pixel' mode random erasing (reference flags reprob/remode/recount,
    template.py:27-33). Operates on the normalized float tensor like timm."""

    def __init__(self, prob=0.25, mode="pixel", count=1,
                 area=(0.02, 1 / 3), aspect=(0.3, 10 / 3)):
        self.prob, self.mode, self.count = prob, mode, count
        self.area, self.aspect = area, aspect

    def __call__(self, t):  # t: float32 HWC tensor
        if np.random.rand() >= self.prob:
            return t
        H, W, C = t.shape
        for _ in range(self.count):
            for _ in range(10):
                area = H * W * np.random.uniform(*self.area)
                ar = np.exp(np.random.uniform(np.log(self.aspect[0]),
                                              np.log(self.aspect[1])))
                h, w = int(round(np.sqrt(area * ar))), int(round(np.sqrt(area / ar)))
                if h < H and w < W:
                    y, x = np.random.randint(0, H - h), np.random.randint(0, W - w)
                    if self.mode == "pixel":
                        t[y:y + h, x:x + w] = torch.randn(h, w, C)
                    else:
                        t[y:y + h, x:x + w] = 0
                    break
        return t


# ------------------------------------------------------------------ full pipelines

class TrainTransform:
    def __init__(self, args, stats_key):
        self.size = args.input_size
        self.small = args.input_size <= 64
        self.color_jitter = getattr(args, "color_jitter", 0.0) or 0.0
        self.randaug = (RandAugment.from_policy(args.aa)
                        if getattr(args, "aa", "") else None)
        mean, std = DATASET_STATS[stats_key]
        self.mean = torch.tensor(mean, dtype=torch.float32)
        self.std = torch.tensor(std, dtype=torch.float32)
        self.erasing = (RandomErasing(args.reprob, args.remode, args.recount)
                        if getattr(args, "reprob", 0) > 0 else None)
        self.disabled = getattr(args, "no_aug", False)

    def __call__(self, img):
        if not self.disabled:
            if self.small:
                img = random_crop_pad(img, self.size, padding=4)
            else:
                img = random_resized_crop(img, self.size)
            if np.random.rand() < 0.5:
                img = img[:, ::-1]
            if self.randaug is not None:
                img = self.randaug(np.ascontiguousarray(img))
            if self.color_jitter > 0:
                img = color_jitter(np.ascontiguousarray(img), self.color_jitter)
        elif img.shape[0] != self.size:
            img = _resize(img, self.size)
        t = torch.from_numpy(np.ascontiguousarray(img)).float().div_(255.0)
        t = (t - self.mean) / self.std
        if self.erasing is not None and not self.disabled:
            t = self.erasing(t)
        return t


class EvalTransform:
    def __init__(self, args, stats_key):
        self.size = args.input_size
        self.small = args.input_size <= 64
        mean, std = DATASET_STATS[stats_key]
        self.mean = torch.tensor(mean, dtype=torch.float32)
        self.std = torch.tensor(std, dtype=torch.float32)

    def __call__(self, img):
        if not self.small:
            scale = int(256 / 224 * self.size)
            short = min(img.shape[0], img.shape[1])
            if short != scale:
                ratio = scale / short
                img = _resize(img, max(int(round(img.shape[0] * ratio)), scale))
            img = _center_crop(img, self.size)
        elif img.shape[0] != self.size:
            img = _resize(img, self.size)
        t = torch.from_numpy(np.ascontiguousarray(img)).float().div_(255.0)
        return (t - self.mean) / self.std
