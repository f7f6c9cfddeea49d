"""ResNet-18/34/50 backbones (NHWC) for the scale-out configs.

The reference repo only ships CIFAR ResNets; BASELINE.json's headline config is
ResNet-18 on CIFAR-100 and ResNet-50 on ImageNet-1000, so these are cilfw-native
additions. CIFAR inputs (<=64 px) get a 3x3/s1 stem with no max-pool; larger inputs
get the classic 7x7/s2 stem + 3x3/s2 max-pool. Projection ("option B") shortcuts:
1x1 conv + BN.
"""

import torch.nn as nn

from ..ops import functional as CF
from .layers import Conv2d, BatchNormAct2d


class BasicBlockB(nn.Module):
    expansion = 1

    def __init__(self, in_ch, out_ch, stride=1):
        super().__init__()
        self.conv1 = Conv2d(in_ch, out_ch, 3, stride=stride, padding=1)
        self.bn1 = BatchNormAct2d(out_ch, relu=True)
        self.conv2 = Conv2d(out_ch, out_ch, 3, stride=1, padding=1)
        self.bn2 = BatchNormAct2d(out_ch, relu=False)
        if stride != 1 or in_ch != out_ch:
            self.proj = Conv2d(in_ch, out_ch, 1, stride=stride, padding=0)
            self.proj_bn = BatchNormAct2d(out_ch, relu=False)
        else:
            self.proj = None

    def forward(self, x):
        out = self.bn1(self.conv1(x))
        residual = self.proj_bn(self.proj(x)) if self.proj is not None else x
        return self.bn2(self.conv2(out), residual)  # fused bn+add+relu


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, in_ch, width, stride=1):
        super().__init__()
        out_ch = width * self.expansion
        self.conv1 = Conv2d(in_ch, width, 1, stride=1, padding=0)
        self.bn1 = BatchNormAct2d(width, relu=True)
        self.conv2 = Conv2d(width, width, 3, stride=stride, padding=1)
        self.bn2 = BatchNormAct2d(width, relu=True)
        self.conv3 = Conv2d(width, out_ch, 1, stride=1, padding=0)
        self.bn3 = BatchNormAct2d(out_ch, relu=False)
        if stride != 1 or in_ch != out_ch:
            self.proj = Conv2d(in_ch, out_ch, 1, stride=stride, padding=0)
            self.proj_bn = BatchNormAct2d(out_ch, relu=False)
        else:
            self.proj = None

    def forward(self, x):
        out = self.bn1(self.conv1(x))
        out = self.bn2(self.conv2(out))
        residual = self.proj_bn(self.proj(x)) if self.proj is not None else x
        return self.bn3(self.conv3(out), residual)  # fused bn+add+relu


class ResNet(nn.Module):
    def __init__(self, block, layers, in_channels=3, small_input=True):
        super().__init__()
        self.small_input = small_input
        if small_input:
            self.stem = Conv2d(in_channels, 64, 3, stride=1, padding=1)
        else:
            self.stem = Conv2d(in_channels, 64, 7, stride=2, padding=3)
        # small_input: stem BN output feeds layer1's first conv AND its
        # identity skip — two consumers, so conv-side BN-backward fusion
        # could never be consumed (with a maxpool stem it is simply unused)
        self.stem_bn = BatchNormAct2d(64, relu=True, fuse_bwd=False)
        self.in_ch = 64
        self.layer1 = self._make_layer(block, 64, layers[0], stride=1)
        self.layer2 = self._make_layer(block, 128, layers[1], stride=2)
        self.layer3 = self._make_layer(block, 256, layers[2], stride=2)
        self.layer4 = self._make_layer(block, 512, layers[3], stride=2)
        self.out_dim = 512 * block.expansion

    def _make_layer(self, block, width, n, stride):
        blocks = [block(self.in_ch, width, stride)]
        self.in_ch = width * block.expansion
        blocks += [block(self.in_ch, width, 1) for _ in range(n - 1)]
        return nn.Sequential(*blocks)

    def forward(self, x):
        x = self.stem_bn(self.stem(x))
        if not self.small_input:
            x = CF.max_pool(x, 3, 2, 1)
        x = self.layer1(x)
        x = self.layer2(x)
        x = self.layer3(x)
        x = self.layer4(x)
        return CF.global_avg_pool(x)


def resnet18(small_input=True, **kw):
    return ResNet(BasicBlockB, [2, 2, 2, 2], small_input=small_input, **kw)


def resnet34(small_input=True, **kw):
    return ResNet(BasicBlockB, [3, 4, 6, 3], small_input=small_input, **kw)


def resnet50(small_input=False, **kw):
    return ResNet(Bottleneck, [3, 4, 6, 3], small_input=small_input, **kw)
