#!/usr/bin/env python3
"""Per-shape conv kernel microbench (GPU): times fwd / bwd-data / bwd-weight for
the ResNet-18-CIFAR layer shapes and prints achieved TFLOP/s.

Usage: python tools/convbench.py [--iters 50] [--shape all|l1|l2|l3|l4|stem]
"""

import argparse
import sys
import os

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402
from cilfw.ops import functional as CF  # noqa: E402

SHAPES = {
    # name: (B, H, C, K, R, stride)
    "stem": (128, 32, 3, 64, 3, 1),
    "l1": (128, 32, 64, 64, 3, 1),
    "l2s": (128, 32, 64, 128, 3, 2),
    "l2": (128, 16, 128, 128, 3, 1),
    "l3s": (128, 16, 128, 256, 3, 2),
    "l3": (128, 8, 256, 256, 3, 1),
    "l4s": (128, 8, 256, 512, 3, 2),
    "l4": (128, 4, 512, 512, 3, 1),
    "proj4": (128, 8, 256, 512, 1, 2),
    "rn50_1x1": (64, 14, 1024, 256, 1, 1),
}


def bench_shape(name, B, H, C, K, R, stride, iters):
    pad = R // 2
    Ho = (H + 2 * pad - R) // stride + 1
    x = torch.randn(B, H, H, C, device="cuda").to(torch.bfloat16)
    x.requires_grad_()
    w = (torch.randn(R, R, C, K, device="cuda") * 0.1).requires_grad_()
    dy = torch.randn(B, Ho, Ho, K, device="cuda").to(torch.bfloat16)

    flops = 2.0 * B * Ho * Ho * K * C * R * R

    def timeit(fn):
        for _ in range(5):
            fn()
        torch.cuda.synchronize()
        t0 = torch.cuda.Event(enable_timing=True)
        t1 = torch.cuda.Event(enable_timing=True)
        t0.record()
        for _ in range(iters):
            fn()
        t1.record()
        torch.cuda.synchronize()
        return t0.elapsed_time(t1) / iters * 1e-3  # seconds

    y = CF.conv2d(x, w, stride, pad)
    tf = timeit(lambda: CF.conv2d(x.detach(), w.detach(), stride, pad))
    g = torch.autograd.grad(y, [x, w], dy, retain_graph=True)

    def bwd():
        torch.autograd.grad(y, [x, w], dy, retain_graph=True)

    tb = timeit(bwd)
    print(f"{name:10s} B{B} {H}x{H} {C:4d}->{K:4d} k{R}s{stride}: "
          f"fwd {tf*1e6:7.1f}us {flops/tf/1e12:6.1f}TF | "
          f"bwd(d+w) {tb*1e6:7.1f}us {2*flops/tb/1e12:6.1f}TF")
    del y, g


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=50)
    ap.add_argument("--shape", default="all")
    args = ap.parse_args()
    names = list(SHAPES) if args.shape == "all" else [args.shape]
    for n in names:
        bench_shape(n, *SHAPES[n], args.iters)
