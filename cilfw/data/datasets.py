"""Dataset sources -> (images uint8 (N,H,W,C), labels int64 (N,)).

The reference pulls CIFAR-100 via continuum's auto-download (utils.py:188-207) and
ImageNet via an ImageFolder adapter (utils.py:171-185). This environment has no
network, so cilfw reads standard on-disk formats when present and otherwise offers a
deterministic synthetic source with the same shapes (used by tests and bench).
"""

import os
import pickle

import numpy as np

try:
    from PIL import Image
except ImportError:  # pragma: no cover
    Image = None

DATASET_STATS = {
    # mean/std in [0,1] per channel
    "cifar100": ((0.5071, 0.4865, 0.4409), (0.2673, 0.2564, 0.2762)),
    "imagenet": ((0.485, 0.456, 0.406), (0.229, 0.224, 0.225)),
    "synthetic": ((0.5, 0.5, 0.5), (0.25, 0.25, 0.25)),
}


def load_cifar100(root, train=True):
    """Reads the standard python pickle format (cifar-100-python/{train,test})."""
    sub = "train" if train else "test"
    path = os.path.join(root, "cifar-100-python", sub)
    if not os.path.exists(path):
        path = os.path.join(root, sub)
    with open(path, "rb") as f:
        d = pickle.load(f, encoding="bytes")
    data = d[b"data"].reshape(-1, 3, 32, 32).transpose(0, 2, 3, 1)  # NHWC uint8
    labels = np.asarray(d[b"fine_labels"], dtype=np.int64)
    return np.ascontiguousarray(data), labels


def make_synthetic(num_classes=100, per_class=50, size=32, seed=0, channels=3):
    """Deterministic class-separable synthetic images: per-class mean color + noise.

    Class means depend only on num_classes (train/eval splits share them; `seed`
    varies the noise), so CIL plumbing tests can verify learning actually
    happens."""
    mean_rng = np.random.default_rng(1234 + num_classes)
    means = mean_rng.integers(40, 216, size=(num_classes, channels))
    rng = np.random.default_rng(seed)
    xs, ys = [], []
    for c in range(num_classes):
        noise = rng.normal(0, 30, size=(per_class, size, size, channels))
        img = np.clip(means[c][None, None, None, :] + noise, 0, 255).astype(np.uint8)
        xs.append(img)
        ys.append(np.full(per_class, c, dtype=np.int64))
    x = np.concatenate(xs)
    y = np.concatenate(ys)
    perm = rng.permutation(len(y))
    return x[perm], y[perm]


def make_synthetic_hard(num_classes=100, per_class=50, size=32, seed=0,
                        channels=3):
    """Low-SNR structured synthetic images — the algorithmic quality gate.

    The separable `make_synthetic` saturates any model at ~100% (it proves
    plumbing, not algorithm quality). Here each class is a random spatial
    TEMPLATE (smooth low-frequency pattern) mixed at low amplitude into
    per-image noise plus a shared distractor pattern, so: accuracy stays
    well below 100%, a 2000-exemplar replay budget is genuinely scarce, and
    catastrophic forgetting / weight-align / KD effects become measurable
    (tools/ablation.py). Templates depend only on (num_classes, size) so
    train/eval share them; `seed` varies noise."""
    trng = np.random.default_rng(5678 + num_classes * 31 + size)
    # classes are MIXTURES of a small shared pattern basis (so classes
    # genuinely overlap), spatially shifted per image (so pixel averaging
    # does not trivially denoise them)
    nbasis = 6
    coarse = trng.normal(0, 1, size=(nbasis, 4, 4, channels))
    reps = size // 4
    basis = np.repeat(np.repeat(coarse, reps, axis=1), reps, axis=2)
    coeff = trng.normal(0, 1, size=(num_classes, nbasis))
    coeff /= np.linalg.norm(coeff, axis=1, keepdims=True)
    templates = np.einsum("cb,bhwk->chwk", coeff, basis)
    distractor = np.repeat(np.repeat(
        trng.normal(0, 1, size=(8, 4, 4, channels)), reps, axis=1),
        reps, axis=2)
    rng = np.random.default_rng(seed)
    xs, ys = [], []
    for c in range(num_classes):
        n = rng.normal(0, 1, size=(per_class, size, size, channels))
        d = distractor[rng.integers(0, 8, per_class)]
        amp = rng.uniform(0.6, 1.4, size=(per_class, 1, 1, 1))
        t = np.broadcast_to(templates[c], (per_class, size, size, channels))
        # per-image random cyclic shift of the class pattern
        sh = rng.integers(0, size, size=(per_class, 2))
        t = np.stack([np.roll(t[i], (sh[i, 0], sh[i, 1]), axis=(0, 1))
                      for i in range(per_class)])
        img = 128 + 30 * (0.40 * amp * t + 0.7 * d + 1.0 * n)
        xs.append(np.clip(img, 0, 255).astype(np.uint8))
        ys.append(np.full(per_class, c, dtype=np.int64))
    x = np.concatenate(xs)
    y = np.concatenate(ys)
    perm = rng.permutation(len(y))
    return x[perm], y[perm]


def load_image(path):
    """Decode one image file -> uint8 HWC RGB (lazy path-based datasets)."""
    with Image.open(path) as im:
        return np.asarray(im.convert("RGB"), dtype=np.uint8)


def scan_imagefolder(root, split):
    """ImageFolder layout (<root>/<split>/<class>/<img>) -> (paths object
    ndarray, labels int64). Lazy: images decode at __getitem__ time, so the
    CIL machinery (scenario splits, add_samples, rehearsal memory) moves only
    path arrays — the reference's continuum ImageFolderDataset contract
    (utils.py:171-185)."""
    base = os.path.join(root, split)
    classes = sorted(d for d in os.listdir(base)
                     if os.path.isdir(os.path.join(base, d)))
    paths, labels = [], []
    for ci, cname in enumerate(classes):
        cdir = os.path.join(base, cname)
        for f in sorted(os.listdir(cdir)):
            if f.lower().endswith((".jpg", ".jpeg", ".png", ".bmp", ".webp")):
                paths.append(os.path.join(cdir, f))
                labels.append(ci)
    return (np.asarray(paths, dtype=object),
            np.asarray(labels, dtype=np.int64), len(classes))


def build_source(args, is_train):
    """-> (x uint8 (N,H,W,C), y int64 (N,), nb_classes, stats_key)."""
    name = args.data_set.lower()
    if name == "cifar100":
        if os.path.exists(os.path.join(args.data_path, "cifar-100-python")) or \
           os.path.exists(os.path.join(args.data_path, "train")):
            x, y = load_cifar100(args.data_path, train=is_train)
        else:
            # offline fallback: synthetic CIFAR-shaped data
            per_class = 500 if is_train else 100
            x, y = make_synthetic(100, per_class, 32,
                                  seed=0 if is_train else 1)
        return x, y, 100, "cifar100"
    if name == "synthetic":
        nc = getattr(args, "synthetic_classes", 100)
        per_class = max(args.synthetic_train_size // nc, 4) if is_train else 10
        x, y = make_synthetic(nc, per_class, args.input_size,
                              seed=0 if is_train else 1)
        return x, y, nc, "synthetic"
    if name == "synthetic_hard":
        nc = getattr(args, "synthetic_classes", 100)
        per_class = max(args.synthetic_train_size // nc, 4) if is_train \
            else 40
        x, y = make_synthetic_hard(nc, per_class, args.input_size,
                                   seed=0 if is_train else 1)
        return x, y, nc, "synthetic"
    if name in ("imagenet100", "imagenet1000", "cub200"):
        nc = {"imagenet100": 100, "imagenet1000": 1000, "cub200": 200}[name]
        split = "train" if is_train else "val"
        if Image is not None and \
                os.path.isdir(os.path.join(args.data_path, split)):
            x, y, found = scan_imagefolder(args.data_path, split)
            return x, y, found, "imagenet"
        per_class = 64 if is_train else 8  # synthetic stand-in (no network)
        x, y = make_synthetic(nc, per_class, args.input_size,
                              seed=0 if is_train else 1)
        return x, y, nc, "imagenet"
    raise NotImplementedError(name)
