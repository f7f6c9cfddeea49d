"""HIP extension loader.

The extension is built IN-TREE (``cilfw/csrc`` → ``cilfw/_hip_ops*.so``) by
``python setup.py build_ext --inplace`` or ``__graft_entry__.build()`` so the binary
travels to GPU boxes with the repo snapshot.

Policy: on a CUDA/ROCm device the HIP kernels are THE compute path — if the extension
is missing we raise instead of silently falling back to ATen (the reference ran
entirely on library kernels; cilfw's point is first-party CDNA4 kernels).
CPU tensors always use the torch reference implementations in ``cilfw/ops/*``.
"""

import importlib
import os

import torch

_ext = None
_tried = False


def _load():
    global _ext, _tried
    if _tried:
        return _ext
    _tried = True
    try:
        _ext = importlib.import_module("cilfw._hip_ops")
    except ImportError:
        _ext = None
    return _ext


def ext():
    """Return the HIP extension module or raise (GPU compute requires it)."""
    e = _load()
    if e is None:
        raise RuntimeError(
            "cilfw._hip_ops extension not built. Run `python setup.py build_ext "
            "--inplace` (or __graft_entry__.build()) — cilfw refuses to run GPU "
            "compute on fallback ATen kernels."
        )
    return e


def have_ext():
    return _load() is not None


def use_hip(t: torch.Tensor) -> bool:
    """True if this tensor should be computed by the HIP kernels."""
    if not t.is_cuda:
        return False
    if os.environ.get("CILFW_FORCE_TORCH") == "1":  # escape hatch for A/B numerics
        return False
    ext()  # raises if missing — no silent fallback on GPU
    return True
