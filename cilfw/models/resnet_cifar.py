"""CIFAR-style ResNet backbones (NHWC), capability-parity with reference resnet.py.

Reference behavior (resnet.py:1-159): 3x3 stem (3->16) + BN + ReLU, three stages of
(depth-2)/6 basic blocks at 16/32/64 channels with strides 1/2/2, parameter-free
"option A" downsample (stride-2 pool + zero-channel pad, resnet.py:9-17), global
avg-pool to a 64-d feature. Returns features only — the classifier lives in
CilModel (reference template.py:87-123).

cilfw differences (deliberate, MI355X-first): NHWC activations, fused BN+ReLU and
add+ReLU epilogues, bf16 compute with fp32 master weights.
"""

import torch
import torch.nn as nn

from ..ops import functional as CF
from .layers import Conv2d, BatchNormAct2d


class BasicBlockA(nn.Module):
    """conv3x3-BN-ReLU-conv3x3-BN + residual + ReLU (reference resnet.py:20-53)."""

    def __init__(self, in_ch, out_ch, stride=1):
        super().__init__()
        self.conv_a = Conv2d(in_ch, out_ch, 3, stride=stride, padding=1)
        self.bn_a = BatchNormAct2d(out_ch, relu=True)
        self.conv_b = Conv2d(out_ch, out_ch, 3, stride=1, padding=1)
        self.bn_b = BatchNormAct2d(out_ch, relu=False)
        self.downsample = (stride != 1 or in_ch != out_ch)
        if self.downsample:
            assert stride == 2 and out_ch == 2 * in_ch, \
                "option-A downsample supports exactly stride2/double-channel"

    def forward(self, x):
        out = self.bn_a(self.conv_a(x))
        residual = CF.downsample_a(x) if self.downsample else x
        return self.bn_b(self.conv_b(out), residual)  # fused bn+add+relu


class CifarResNet(nn.Module):
    """depth in {20,32,44,56,110}; out_dim = 64 (reference resnet.py:80)."""

    def __init__(self, depth=32, in_channels=3):
        super().__init__()
        assert (depth - 2) % 6 == 0, "depth must be 6n+2"
        n = (depth - 2) // 6
        self.depth = depth
        self.conv_1_3x3 = Conv2d(in_channels, 16, 3, stride=1, padding=1)
        self.bn_1 = BatchNormAct2d(16, relu=True, fuse_bwd=False)  # feeds
        # both stage_1's first conv AND its identity skip
        self.stage_1 = self._make_stage(16, 16, n, stride=1)
        self.stage_2 = self._make_stage(16, 32, n, stride=2)
        self.stage_3 = self._make_stage(32, 64, n, stride=2)
        self.out_dim = 64

    @staticmethod
    def _make_stage(in_ch, out_ch, n, stride):
        blocks = [BasicBlockA(in_ch, out_ch, stride)]
        blocks += [BasicBlockA(out_ch, out_ch, 1) for _ in range(n - 1)]
        return nn.Sequential(*blocks)

    def forward(self, x):
        """x: (N,H,W,C) NHWC -> (N, out_dim) features."""
        x = self.bn_1(self.conv_1_3x3(x))
        x = self.stage_1(x)
        x = self.stage_2(x)
        x = self.stage_3(x)
        return CF.global_avg_pool(x)


def resnet20(**kw):
    return CifarResNet(20, **kw)


def resnet32(**kw):
    return CifarResNet(32, **kw)


def resnet44(**kw):
    return CifarResNet(44, **kw)


def resnet56(**kw):
    return CifarResNet(56, **kw)


def resnet110(**kw):
    return CifarResNet(110, **kw)
