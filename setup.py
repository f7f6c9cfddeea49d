"""Build the cilfw HIP extension IN-TREE: cilfw/_hip_lib.so (gfx950 only).

Plain hipcc — the kernels have no torch C++ dependency (the Python side talks to
them through ctypes with raw device pointers + the current HIP stream), so there
is no hipify step and no torch ABI coupling anywhere.

    python setup.py build_ext --inplace
"""

import os
import subprocess
import sys

REPO = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(REPO, "cilfw", "csrc")
OUT = os.path.join(REPO, "cilfw", "_hip_lib.so")
SOURCES = ["conv.hip", "norm.hip", "gemm.hip", "loss.hip", "util.hip"]
HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
ARCH = os.environ.get("CILFW_GPU_ARCH", "gfx950")


def _mtime(p):
    return os.path.getmtime(p) if os.path.exists(p) else 0.0


def build(force=False):
    hdr = os.path.join(CSRC, "common.h")
    newest_src = max(_mtime(os.path.join(CSRC, s)) for s in SOURCES)
    newest_src = max(newest_src, _mtime(hdr), _mtime(os.path.abspath(__file__)))
    if not force and _mtime(OUT) > newest_src:
        print(f"cilfw/_hip_lib.so up to date ({OUT})")
        return OUT
    objs = []
    os.makedirs(os.path.join(REPO, "build"), exist_ok=True)
    for s in SOURCES:
        src = os.path.join(CSRC, s)
        obj = os.path.join(REPO, "build", s.replace(".hip", ".o"))
        cmd = [HIPCC, f"--offload-arch={ARCH}", "-O3", "-std=c++17", "-fPIC",
               "-I", CSRC, "-c", src, "-o", obj]
        print(" ".join(cmd))
        subprocess.run(cmd, check=True)
        objs.append(obj)
    cmd = [HIPCC, f"--offload-arch={ARCH}", "-shared", "-fPIC", "-o", OUT] + objs
    print(" ".join(cmd))
    subprocess.run(cmd, check=True)
    return OUT


if __name__ == "__main__":
    force = "--force" in sys.argv
    if "build_ext" in sys.argv or "build" in sys.argv or force or \
            len(sys.argv) == 1:
        build(force=force)
    else:
        print(f"usage: {sys.argv[0]} build_ext --inplace [--force]")
