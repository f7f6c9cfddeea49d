"""nn.Module wrappers over cilfw ops (NHWC activations, fp32 master params)."""

import math

import torch
import torch.nn as nn

from ..ops import functional as CF


class Conv2d(nn.Module):
    """Bias-free conv, weight stored (R,S,C,K) fp32 (implicit-GEMM B-operand layout).

    He-normal init like the reference (resnet.py:84-86)."""

    def __init__(self, in_ch, out_ch, kernel_size, stride=1, padding=0):
        super().__init__()
        self.in_ch, self.out_ch = in_ch, out_ch
        self.kernel_size, self.stride, self.padding = kernel_size, stride, padding
        k = kernel_size
        self.weight = nn.Parameter(torch.empty(k, k, in_ch, out_ch))
        fan_in = k * k * in_ch
        nn.init.normal_(self.weight, 0.0, math.sqrt(2.0 / fan_in))

    def forward(self, x):
        return CF.conv2d(x, self.weight, self.stride, self.padding)

    def extra_repr(self):
        return (f"{self.in_ch}, {self.out_ch}, k={self.kernel_size}, "
                f"s={self.stride}, p={self.padding}")


class BatchNormAct2d(nn.Module):
    """BN over NHWC with optional fused ReLU. gamma=1, beta=0 init
    (reference resnet.py:88-90)."""

    def __init__(self, num_features, eps=1e-5, momentum=0.1, relu=False,
                 fuse_bwd=True):
        super().__init__()
        self.num_features, self.eps, self.momentum, self.relu = (
            num_features, eps, momentum, relu)
        # fuse_bwd=False: this BN's output has MULTIPLE consumers (e.g. a
        # stem feeding both a conv and an identity skip) — the downstream
        # conv's bwd-data epilogue must not waste work emitting BN-backward
        # partials that can never be consumed (the accumulated grad tensor
        # loses the attachment). Single-consumer block BNs keep the default.
        self.fuse_bwd = fuse_bwd
        self.weight = nn.Parameter(torch.ones(num_features))
        self.bias = nn.Parameter(torch.zeros(num_features))
        self.register_buffer("running_mean", torch.zeros(num_features))
        self.register_buffer("running_var", torch.ones(num_features))

    def forward(self, x, residual=None):
        # A frozen teacher's (mean, invstd) are precomputed once
        # (CilModel.cast_compute_weights_); that is only valid while the
        # running stats never change — a train-mode forward would update them
        # and silently desynchronize the cached pair.
        assert not (self.training
                    and getattr(self.running_mean, "_cilfw_frozen", None)
                    is not None), \
            "BN has precomputed frozen eval stats but is in train mode"
        if residual is not None:
            # fused y = relu(bn(x) + residual) — residual-block tail
            return CF.batchnorm_add_relu(x, residual, self.weight, self.bias,
                                         self.running_mean, self.running_var,
                                         self.momentum, self.eps,
                                         self.training)
        return CF.batchnorm_act(x, self.weight, self.bias, self.running_mean,
                                self.running_var, self.momentum, self.eps,
                                self.training, self.relu,
                                fuse_bwd=self.fuse_bwd)

    def extra_repr(self):
        return f"{self.num_features}, relu={self.relu}"


class Linear(nn.Module):
    def __init__(self, in_features, out_features, bias=True):
        super().__init__()
        self.in_features, self.out_features = in_features, out_features
        self.weight = nn.Parameter(torch.empty(out_features, in_features))
        nn.init.kaiming_normal_(self.weight)  # reference resnet.py:87 style
        if bias:
            self.bias = nn.Parameter(torch.zeros(out_features))
        else:
            self.register_parameter("bias", None)

    def forward(self, x):
        return CF.linear(x, self.weight, self.bias)

    def extra_repr(self):
        return f"{self.in_features}, {self.out_features}"
