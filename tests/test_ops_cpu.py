"""CPU reference-op correctness: cilfw autograd Functions vs torch autograd.

These CPU implementations are the numerics oracles the HIP kernels are tested
against (tests/test_ops_gpu.py), so they must themselves match torch.
"""

import torch
import torch.nn.functional as F
import pytest

from cilfw import ops


def _nchw(x):
    return x.permute(0, 3, 1, 2).contiguous()


@pytest.mark.parametrize("cin,cout,k,stride,hw", [
    (8, 16, 3, 1, 8), (16, 32, 3, 2, 8), (8, 8, 1, 1, 6), (3, 16, 3, 1, 8),
    (8, 16, 1, 2, 8), (3, 16, 7, 2, 16),
])
def test_conv_matches_torch(cin, cout, k, stride, hw):
    torch.manual_seed(0)
    pad = k // 2
    x = torch.randn(2, hw, hw, cin, requires_grad=True)
    w = (torch.randn(k, k, cin, cout) * 0.1).requires_grad_()
    y = ops.conv2d(x, w, stride, pad)
    # torch oracle
    xt = _nchw(x.detach()).requires_grad_(True)
    wt = w.detach().permute(3, 2, 0, 1).contiguous().requires_grad_(True)
    yt = F.conv2d(xt, wt, stride=stride, padding=pad)
    assert torch.allclose(y, yt.permute(0, 2, 3, 1), atol=1e-4, rtol=1e-4)
    dy = torch.randn_like(y)
    y.backward(dy)
    yt.backward(_nchw(dy))
    assert torch.allclose(x.grad, xt.grad.permute(0, 2, 3, 1), atol=1e-4,
                          rtol=1e-4)
    assert torch.allclose(w.grad, wt.grad.permute(2, 3, 1, 0), atol=1e-4,
                          rtol=1e-4)


def test_bn_matches_torch_training():
    torch.manual_seed(0)
    C = 8
    x = torch.randn(4, 5, 5, C, requires_grad=True)
    g = (torch.rand(C) + 0.5).requires_grad_()
    b = torch.randn(C, requires_grad=True)
    rm, rv = torch.zeros(C), torch.ones(C)
    y = ops.batchnorm_act(x, g, b, rm, rv, momentum=0.1, training=True,
                          relu=False)
    # torch oracle
    xt = _nchw(x.detach()).requires_grad_(True)
    bn = torch.nn.BatchNorm2d(C, momentum=0.1)
    with torch.no_grad():
        bn.weight.copy_(g)
        bn.bias.copy_(b)
    yt = bn(xt)
    assert torch.allclose(y, yt.permute(0, 2, 3, 1), atol=1e-5, rtol=1e-4)
    assert torch.allclose(rm, bn.running_mean, atol=1e-6)
    assert torch.allclose(rv, bn.running_var, atol=1e-5)
    dy = torch.randn_like(y)
    y.backward(dy)
    yt.backward(_nchw(dy))
    assert torch.allclose(x.grad, xt.grad.permute(0, 2, 3, 1), atol=1e-5,
                          rtol=1e-4)
    assert torch.allclose(g.grad, bn.weight.grad, atol=1e-4)
    assert torch.allclose(b.grad, bn.bias.grad, atol=1e-4)


def test_bn_relu_fused():
    torch.manual_seed(1)
    C = 8
    x = torch.randn(4, 5, 5, C, requires_grad=True)
    g = torch.rand(C) + 0.5
    b = torch.randn(C)
    rm, rv = torch.zeros(C), torch.ones(C)
    y = ops.batchnorm_act(x, g, b, rm, rv, training=True, relu=True)
    assert (y >= 0).all()
    y.sum().backward()
    assert torch.isfinite(x.grad).all()


def test_bn_eval_mode():
    C = 4
    x = torch.randn(2, 3, 3, C)
    g, b = torch.ones(C), torch.zeros(C)
    rm, rv = torch.randn(C) * 0.1, torch.rand(C) + 0.5
    y = ops.batchnorm_act(x, g, b, rm, rv, training=False)
    expected = (x - rm) / torch.sqrt(rv + 1e-5)
    assert torch.allclose(y, expected, atol=1e-5)


def test_add_relu_and_downsample():
    a = torch.randn(2, 4, 4, 8, requires_grad=True)
    b = torch.randn(2, 4, 4, 8, requires_grad=True)
    y = ops.add_relu(a, b)
    assert torch.allclose(y, (a + b).clamp_min(0), atol=1e-6)
    y.sum().backward()
    assert torch.allclose(a.grad, ((a + b) > 0).float(), atol=1e-6)

    x = torch.randn(2, 4, 4, 8, requires_grad=True)
    d = ops.downsample_a(x)
    assert d.shape == (2, 2, 2, 16)
    assert torch.equal(d[..., :8], x.detach()[:, ::2, ::2, :])
    assert (d[..., 8:] == 0).all()
    d.sum().backward()
    assert x.grad[:, ::2, ::2, :].eq(1).all()
    assert x.grad[:, 1::2, :, :].eq(0).all()


def test_gap_and_linear():
    x = torch.randn(3, 4, 4, 8, requires_grad=True)
    y = ops.global_avg_pool(x)
    assert torch.allclose(y, x.mean(dim=(1, 2)), atol=1e-6)
    y.sum().backward()
    assert torch.allclose(x.grad, torch.full_like(x, 1 / 16.0))

    xx = torch.randn(5, 8, requires_grad=True)
    w = torch.randn(3, 8, requires_grad=True)
    b = torch.randn(3, requires_grad=True)
    out = ops.linear(xx, w, b)
    ref = xx.detach() @ w.detach().t() + b.detach()
    assert torch.allclose(out, ref, atol=1e-5)
    dy = torch.randn_like(out)
    out.backward(dy)
    assert torch.allclose(xx.grad, dy @ w.detach(), atol=1e-5)
    assert torch.allclose(w.grad, dy.t() @ xx.detach(), atol=1e-5)
    assert torch.allclose(b.grad, dy.sum(0), atol=1e-5)


def test_maxpool_matches_torch():
    torch.manual_seed(0)
    x = torch.randn(2, 8, 8, 4, requires_grad=True)
    y = ops.max_pool(x, 3, 2, 1)
    xt = _nchw(x.detach()).requires_grad_(True)
    yt = F.max_pool2d(xt, 3, 2, 1)
    assert torch.allclose(y, yt.permute(0, 2, 3, 1), atol=1e-6)
    dy = torch.randn_like(y)
    y.backward(dy)
    yt.backward(_nchw(dy))
    assert torch.allclose(x.grad, xt.grad.permute(0, 2, 3, 1), atol=1e-6)


@pytest.mark.parametrize("smooth", [0.0, 0.1])
def test_cross_entropy_matches_torch(smooth):
    torch.manual_seed(0)
    logits = torch.randn(7, 11, requires_grad=True)
    targets = torch.randint(0, 11, (7,))
    loss = ops.cross_entropy(logits, targets, smooth)
    lt = logits.detach().requires_grad_(True)
    ref = F.cross_entropy(lt, targets, label_smoothing=smooth)
    assert torch.allclose(loss, ref, atol=1e-6)
    loss.backward()
    ref.backward()
    assert torch.allclose(logits.grad, lt.grad, atol=1e-6)


def test_kd_matches_reference_formula():
    """SoftTarget (reference utils.py:121-132): KLDiv(log_softmax(s/T),
    softmax(t/T), batchmean) * T*T."""
    torch.manual_seed(0)
    T = 2.0
    s = torch.randn(5, 9, requires_grad=True)
    t = torch.randn(5, 9)
    loss = ops.kd_loss(s, t, T)
    st = s.detach().requires_grad_(True)
    ref = F.kl_div(F.log_softmax(st / T, dim=1), F.softmax(t / T, dim=1),
                   reduction="batchmean") * T * T
    assert torch.allclose(loss, ref, atol=1e-6)
    loss.backward()
    ref.backward()
    assert torch.allclose(s.grad, st.grad, atol=1e-6)


def test_accuracy_topk():
    logits = torch.tensor([[0.1, 0.9, 0.0], [0.8, 0.1, 0.1], [0.2, 0.3, 0.5]])
    targets = torch.tensor([1, 1, 2])
    a1, a2 = ops.accuracy(logits, targets, topk=(1, 2))
    assert abs(a1 - 200.0 / 3) < 1e-6
    assert abs(a2 - 100.0) < 1e-6


def test_bn_add_relu_fused_matches_composite():
    torch.manual_seed(2)
    C = 8
    x = torch.randn(4, 5, 5, C, requires_grad=True)
    res = torch.randn(4, 5, 5, C, requires_grad=True)
    g = (torch.rand(C) + 0.5).requires_grad_()
    b = torch.randn(C, requires_grad=True)
    rm, rv = torch.zeros(C), torch.ones(C)
    y = ops.batchnorm_add_relu(x, res, g, b, rm, rv, training=True)
    # composite oracle
    x2 = x.detach().requires_grad_()
    res2 = res.detach().requires_grad_()
    g2 = g.detach().requires_grad_()
    b2 = b.detach().requires_grad_()
    rm2, rv2 = torch.zeros(C), torch.ones(C)
    y2 = ops.add_relu(ops.batchnorm_act(x2, g2, b2, rm2, rv2, training=True),
                      res2)
    assert torch.allclose(y, y2, atol=1e-5)
    assert torch.allclose(rm, rm2, atol=1e-7)
    dy = torch.randn_like(y)
    y.backward(dy)
    y2.backward(dy)
    for a, c in [(x, x2), (res, res2), (g, g2), (b, b2)]:
        assert torch.allclose(a.grad, c.grad, atol=1e-5)


def test_wa_loss_matches_composite():
    torch.manual_seed(4)
    M, C, Ck = 9, 20, 12
    s = torch.randn(M, C, requires_grad=True)
    t = torch.randn(M, Ck)
    y = torch.randint(0, C, (M,))
    lam, T, sm = 0.5, 2.0, 0.1
    total, ce, kd = ops.wa_loss(s, t, y, sm, T, lam)
    s2 = s.detach().requires_grad_()
    ce2 = F.cross_entropy(s2, y, label_smoothing=sm)
    kd2 = F.kl_div(F.log_softmax(s2[:, :Ck] / T, 1),
                   F.softmax(t / T, 1), reduction="batchmean") * T * T
    tot2 = ce2 + lam * kd2
    assert torch.allclose(total, tot2, atol=1e-6)
    assert torch.allclose(ce, ce2, atol=1e-6)
    assert torch.allclose(kd, kd2, atol=1e-6)
    total.backward()
    tot2.backward()
    assert torch.allclose(s.grad, s2.grad, atol=1e-6)


def test_wa_loss_no_teacher():
    torch.manual_seed(5)
    s = torch.randn(7, 11, requires_grad=True)
    y = torch.randint(0, 11, (7,))
    total, ce, kd = ops.wa_loss(s, None, y, 0.0, 2.0, 0.5)
    assert kd.item() == 0.0
    ref = F.cross_entropy(s.detach(), y)
    assert torch.allclose(total, ref, atol=1e-6)
    total.backward()
    assert torch.isfinite(s.grad).all()
