"""CilModel / CilClassifier — dynamic multi-head classifier + WA weight alignment.

Capability parity with reference template.py:87-166:
- CilClassifier: one linear head appended per task (adaption, template.py:103-104);
  forward = concatenation of all head logits in head order (template.py:99-101).
  cilfw computes it as ONE fused GEMM over the row-concatenated head weights instead
  of the reference's per-head GEMM + torch.cat.
- CilModel.forward -> (logits, features) (template.py:120-123); extract_vector
  (:117); copy (:125); freeze (:128-144); prev_model_adaption (:146-150);
  after_model_adaption -> weight_align (:152-166).
- weight_align: gamma = mean(||w_old rows||) / mean(||w_new rows||), rescales the
  NEWEST head's weight in place (the core WA step, template.py:156-166).
"""

import copy as _copy

import torch
import torch.nn as nn

from ..ops import functional as CF
from . import resnet_cifar, resnet


def get_backbone(name, input_size=32):
    """Backbone factory (reference template.py:72-84 + BASELINE scale-out models)."""
    small = input_size <= 64
    factories = {
        "resnet20": lambda: resnet_cifar.resnet20(),
        "resnet32": lambda: resnet_cifar.resnet32(),
        "resnet44": lambda: resnet_cifar.resnet44(),
        "resnet56": lambda: resnet_cifar.resnet56(),
        "resnet110": lambda: resnet_cifar.resnet110(),
        "resnet18": lambda: resnet.resnet18(small_input=small),
        "resnet34": lambda: resnet.resnet34(small_input=small),
        "resnet50": lambda: resnet.resnet50(small_input=small),
    }
    if name not in factories:
        raise NotImplementedError(f"unknown backbone {name!r}")
    return factories[name]()


class CilClassifier(nn.Module):
    """Growable multi-head linear classifier; logits = [head_0 | head_1 | ...]."""

    def __init__(self, embed_dim, nb_classes):
        super().__init__()
        self.embed_dim = embed_dim
        self.heads = nn.ModuleList()
        self.adaption(nb_classes)

    def adaption(self, nb_new_classes):
        head = nn.Module()
        head.weight = nn.Parameter(torch.empty(nb_new_classes, self.embed_dim))
        nn.init.kaiming_normal_(head.weight)
        head.bias = nn.Parameter(torch.zeros(nb_new_classes))
        head.out_features = nb_new_classes
        self.heads.append(head)

    @property
    def nb_classes(self):
        return sum(h.out_features for h in self.heads)

    def forward(self, x):
        # one fused GEMM over concatenated head weights (heads are tiny:
        # C x 64..2048); frozen models use a cached concatenation
        cat = getattr(self, "_frozen_cat", None)
        if cat is not None:
            return CF.linear(x, cat[0], cat[1])
        w = torch.cat([h.weight for h in self.heads], dim=0)
        b = torch.cat([h.bias for h in self.heads], dim=0)
        return CF.linear(x, w, b)

    def __getitem__(self, i):
        return self.heads[i]

    def __len__(self):
        return len(self.heads)


class CilModel(nn.Module):
    def __init__(self, backbone_name, input_size=32):
        super().__init__()
        self.backbone = get_backbone(backbone_name, input_size)
        self.fc = None

    @property
    def feature_dim(self):
        return self.backbone.out_dim

    def extract_vector(self, x):
        return self.backbone(x)

    def forward(self, x):
        features = self.backbone(x)
        logits = self.fc(features)
        return logits, features

    def copy(self):
        return _copy.deepcopy(self)

    def freeze(self, names=("all",)):
        """Freeze submodules by name; 'all' freezes everything and sets eval
        (reference template.py:128-144)."""
        freezed = []
        for name in names:
            if name == "all":
                for p in self.parameters():
                    p.requires_grad = False
                self.eval()
                return ["all"]
            if hasattr(self, name):
                freezed.append(name)
                mod = getattr(self, name)
                for p in mod.parameters():
                    p.requires_grad = False
                mod.eval()
        missing = set(names) - set(freezed)
        if missing:
            raise AttributeError(f"unknown submodules to freeze: {missing}")
        return freezed

    @torch.no_grad()
    def cast_compute_weights_(self, dtype):
        """Convert conv/linear weights to the compute dtype in place — used on
        the FROZEN teacher so its forward skips the per-step fp32->bf16 casts
        (the fp32 masters only matter for models that train). Also precomputes
        each BN's eval (mean, invstd) so the per-call finalize launch is
        skipped (the running stats of a frozen model never change)."""
        from .layers import Conv2d, BatchNormAct2d
        for m in self.modules():
            if isinstance(m, Conv2d):
                m.weight.data = m.weight.data.to(dtype)
            elif isinstance(m, BatchNormAct2d):
                mean = m.running_mean.float().clone()
                invstd = torch.rsqrt(m.running_var.float() + m.eps)
                m.running_mean._cilfw_frozen = (mean.contiguous(),
                                                invstd.contiguous())
        if self.fc is not None:
            for h in self.fc.heads:
                h.weight.data = h.weight.data.to(dtype)
                h.bias.data = h.bias.data.to(dtype)
            self.fc._frozen_cat = (
                torch.cat([h.weight for h in self.fc.heads], 0).contiguous(),
                torch.cat([h.bias for h in self.fc.heads], 0).contiguous())
        return self

    def prev_model_adaption(self, nb_classes):
        if self.fc is None:
            self.fc = CilClassifier(self.feature_dim, nb_classes)
        else:
            self.fc.adaption(nb_classes)
        # keep the new head on the model's device/dtype
        ref = next(self.backbone.parameters())
        self.fc.to(ref.device)

    def after_model_adaption(self, nb_classes, args=None):
        task_id = getattr(args, "task_id", len(self.fc) - 1) if args is not None \
            else len(self.fc) - 1
        if task_id > 0:
            self.weight_align(nb_classes)

    @torch.no_grad()
    def weight_align(self, nb_new_classes):
        """WA: rescale the newest head so mean new-row norm == mean old-row norm."""
        w = torch.cat([h.weight for h in self.fc.heads], dim=0)
        norms = torch.norm(w, p=2, dim=1)
        norm_old = norms[:-nb_new_classes]
        norm_new = norms[-nb_new_classes:]
        gamma = norm_old.mean() / norm_new.mean()
        print(f"old norm: {norm_old.mean().item():.4f}, "
              f"new norm: {norm_new.mean().item():.4f}, gamma: {gamma.item():.4f}")
        self.fc.heads[-1].weight.mul_(gamma)
        return gamma.item()


def freeze_parameters(m, requires_grad=False):
    """reference template.py:61-69."""
    if isinstance(m, nn.Parameter):
        m.requires_grad = requires_grad
    else:
        for p in m.parameters():
            p.requires_grad = requires_grad
