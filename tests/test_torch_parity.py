"""Compositional parity: a cilfw CifarResNet trained with FlatSGD must track a
weight-identical torch-native (nn.Conv2d/BatchNorm2d/SGD) model step for step.

The per-op oracles (test_ops_cpu) verify each kernel's math; this guards the
COMPOSITION — layout conversions, flat-param rebinding, fused SGD, cosine LR —
against the reference-equivalent torch stack."""

import torch
import torch.nn as nn
import torch.nn.functional as F

from cilfw.models import CilModel
from cilfw.distributed.ddp import DataParallelEngine
from cilfw.optim import FlatSGD
from cilfw import ops


class TorchBlock(nn.Module):
    def __init__(self, cin, cout, stride):
        super().__init__()
        self.conv_a = nn.Conv2d(cin, cout, 3, stride, 1, bias=False)
        self.bn_a = nn.BatchNorm2d(cout)
        self.conv_b = nn.Conv2d(cout, cout, 3, 1, 1, bias=False)
        self.bn_b = nn.BatchNorm2d(cout)
        self.down = stride != 1 or cin != cout

    def forward(self, x):
        out = F.relu(self.bn_a(self.conv_a(x)))
        out = self.bn_b(self.conv_b(out))
        if self.down:
            r = F.avg_pool2d(x, 1, 2)
            r = torch.cat([r, r.mul(0)], 1)
        else:
            r = x
        return F.relu(out + r)


class TorchCifarNet(nn.Module):
    """NCHW mirror of cilfw's CifarResNet-20 + 1-head classifier."""

    def __init__(self, nc):
        super().__init__()
        self.stem = nn.Conv2d(3, 16, 3, 1, 1, bias=False)
        self.bn = nn.BatchNorm2d(16)
        blocks = []
        for cin, cout, stride, n in [(16, 16, 1, 3), (16, 32, 2, 3),
                                     (32, 64, 2, 3)]:
            blocks.append(TorchBlock(cin, cout, stride))
            blocks += [TorchBlock(cout, cout, 1) for _ in range(n - 1)]
        self.blocks = nn.Sequential(*blocks)
        self.fc = nn.Linear(64, nc)

    def forward(self, x):
        x = F.relu(self.bn(self.stem(x)))
        x = self.blocks(x)
        x = x.mean(dim=(2, 3))
        return self.fc(x)


def _copy_weights(cil, tm):
    """cilfw (R,S,C,K) convs / heads -> torch mirror."""
    def conv_w(m):
        return m.weight.detach().permute(3, 2, 0, 1).contiguous()

    bb = cil.backbone
    tm.stem.weight.data.copy_(conv_w(bb.conv_1_3x3))
    tm.bn.weight.data.copy_(bb.bn_1.weight)
    tm.bn.bias.data.copy_(bb.bn_1.bias)
    cil_blocks = (list(bb.stage_1) + list(bb.stage_2) + list(bb.stage_3))
    for cb, tb in zip(cil_blocks, tm.blocks):
        tb.conv_a.weight.data.copy_(conv_w(cb.conv_a))
        tb.conv_b.weight.data.copy_(conv_w(cb.conv_b))
        tb.bn_a.weight.data.copy_(cb.bn_a.weight)
        tb.bn_a.bias.data.copy_(cb.bn_a.bias)
        tb.bn_b.weight.data.copy_(cb.bn_b.weight)
        tb.bn_b.bias.data.copy_(cb.bn_b.bias)
    tm.fc.weight.data.copy_(cil.fc.heads[0].weight)
    tm.fc.bias.data.copy_(cil.fc.heads[0].bias)


def test_training_tracks_torch_native():
    torch.manual_seed(0)
    nc = 7
    cil = CilModel("resnet20", 32)
    cil.prev_model_adaption(nc)
    tm = TorchCifarNet(nc)
    _copy_weights(cil, tm)

    engine = DataParallelEngine(cil)
    opt = FlatSGD(engine, lr=0.05, momentum=0.9, weight_decay=5e-4)
    topt = torch.optim.SGD(tm.parameters(), lr=0.05, momentum=0.9,
                           weight_decay=5e-4)

    g = torch.Generator().manual_seed(11)
    for step in range(5):
        x = torch.randn(8, 32, 32, 3, generator=g)
        y = torch.randint(0, nc, (8,), generator=g)

        opt.zero_grad()
        logits, _ = cil(x)
        loss = ops.cross_entropy(logits.float(), y)
        loss.backward()
        engine.finalize()
        opt.step()

        topt.zero_grad()
        tl = tm(x.permute(0, 3, 1, 2).contiguous())
        tloss = F.cross_entropy(tl, y)
        tloss.backward()
        topt.step()

        assert torch.allclose(loss, tloss, atol=1e-4, rtol=1e-4), \
            f"step {step}: loss diverged {loss.item()} vs {tloss.item()}"
        assert torch.allclose(logits, tl, atol=5e-3, rtol=1e-3), \
            f"step {step}: logits diverged"

    # final weights still aligned
    w_cil = cil.backbone.conv_1_3x3.weight.detach().permute(3, 2, 0, 1)
    assert torch.allclose(w_cil, tm.stem.weight.detach(), atol=2e-3), \
        "weights drifted beyond fp-reordering tolerance"
