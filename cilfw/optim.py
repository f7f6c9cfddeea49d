"""Fused SGD + cosine LR — the reference recipe as flat-buffer single-kernel ops.

Reference: torch.optim.SGD(lr, momentum, weight_decay) + CosineAnnealingLR(T_max)
(template.py:246-249, stepped per epoch :278). cilfw runs the whole update as ONE
kernel over the DataParallelEngine's flat param/grad buffers (HIP kernel
``sgd_step`` on GPU; two fused ATen ops on CPU):

    g = grad + wd * p;  m = mu * m + g;  p -= lr * m
"""

import math

import torch

from .ops._backend import use_hip, ext


class FlatSGD:
    def __init__(self, engine, lr, momentum=0.9, weight_decay=5e-4):
        self.engine = engine
        # lr lives in a 1-elem DEVICE tensor on GPU: the hipGraph-captured
        # step reads it through memory, so the cosine scheduler only writes
        # the scalar per epoch instead of forcing a per-epoch re-capture
        self._lr_dev = (torch.full((1,), float(lr), dtype=torch.float32,
                                   device=engine.flat_params.device)
                        if engine.flat_params.is_cuda else None)
        self._lr = float(lr)
        self.base_lr = lr
        self.momentum = momentum
        self.weight_decay = weight_decay
        self.momentum_buf = torch.zeros_like(engine.flat_grads)

    @property
    def lr(self):
        return self._lr

    @lr.setter
    def lr(self, v):
        self._lr = float(v)
        if self._lr_dev is not None:
            self._lr_dev.fill_(float(v))

    @torch.no_grad()
    def step(self):
        p, g, m = (self.engine.flat_params, self.engine.flat_grads,
                   self.momentum_buf)
        mirror = getattr(self.engine, "flat_bf16", None)
        if use_hip(p):
            ext().sgd_step(p, g, m, self._lr_dev, self.momentum,
                           self.weight_decay, mirror)
            return
        # CPU reference (and CILFW_FORCE_TORCH): same math, fused ATen ops
        if self.weight_decay != 0:
            g = g.add(p, alpha=self.weight_decay)
        m.mul_(self.momentum).add_(g)
        p.add_(m, alpha=-self.lr)
        if mirror is not None:
            mirror.copy_(p)

    def zero_grad(self):
        self.engine.zero_grad()

    def state_dict(self):
        return {"lr": self.lr, "base_lr": self.base_lr,
                "momentum": self.momentum, "weight_decay": self.weight_decay,
                "momentum_buf": self.momentum_buf}

    def load_state_dict(self, sd):
        self.lr = sd["lr"]
        self.base_lr = sd["base_lr"]
        self.momentum = sd["momentum"]
        self.weight_decay = sd["weight_decay"]
        self.momentum_buf.copy_(sd["momentum_buf"])


class CosineLR:
    """CosineAnnealingLR(T_max) equivalent, stepped once per epoch."""

    def __init__(self, optimizer, t_max, eta_min=0.0):
        self.opt = optimizer
        self.t_max = t_max
        self.eta_min = eta_min
        self.epoch = 0

    def step(self):
        self.epoch += 1
        self.opt.lr = self.eta_min + (self.opt.base_lr - self.eta_min) * \
            (1 + math.cos(math.pi * min(self.epoch, self.t_max) / self.t_max)) / 2

    def state_dict(self):
        return {"epoch": self.epoch, "t_max": self.t_max, "eta_min": self.eta_min}

    def load_state_dict(self, sd):
        self.epoch = sd["epoch"]
        self.t_max = sd["t_max"]
        self.eta_min = sd["eta_min"]
