import torch

from cilfw.models import CilModel, get_backbone


def test_cifar_backbone_shapes():
    m = get_backbone("resnet20", 32)
    x = torch.randn(2, 32, 32, 3)
    f = m(x)
    assert f.shape == (2, 64)
    assert m.out_dim == 64


def test_resnet18_cifar_shapes():
    m = get_backbone("resnet18", 32)
    f = m(torch.randn(2, 32, 32, 3))
    assert f.shape == (2, 512)


def test_resnet50_imagenet_shapes():
    m = get_backbone("resnet50", 224)
    f = m(torch.randn(1, 64, 64, 3))  # any spatial size works w/ GAP
    assert f.shape == (1, 2048)


def test_resnet32_param_count():
    """Structural parity with the reference CifarResNet-32 (resnet.py:56-91):
    analytic conv+BN parameter count for depth 32 (n=5 blocks/stage)."""
    m = get_backbone("resnet32", 32)
    n = sum(p.numel() for p in m.parameters())

    def conv(cin, cout):
        return 9 * cin * cout

    def bn(c):
        return 2 * c

    expected = conv(3, 16) + bn(16)
    for cin, cout, nblk in [(16, 16, 5), (16, 32, 5), (32, 64, 5)]:
        expected += conv(cin, cout) + bn(cout) + conv(cout, cout) + bn(cout)
        expected += (nblk - 1) * (2 * conv(cout, cout) + 2 * bn(cout))
    assert n == expected


def test_head_growth_and_forward():
    model = CilModel("resnet20", 32)
    model.prev_model_adaption(5)
    logits, feats = model(torch.randn(3, 32, 32, 3))
    assert logits.shape == (3, 5)
    model.prev_model_adaption(7)
    logits, _ = model(torch.randn(3, 32, 32, 3))
    assert logits.shape == (3, 12)
    assert len(model.fc) == 2
    assert model.fc[1].out_features == 7


def test_head_concat_order():
    """logits columns = head order (reference template.py:99-101)."""
    model = CilModel("resnet20", 32)
    model.prev_model_adaption(3)
    model.prev_model_adaption(2)
    with torch.no_grad():
        model.fc.heads[0].weight.zero_()
        model.fc.heads[0].bias.fill_(1.0)
        model.fc.heads[1].weight.zero_()
        model.fc.heads[1].bias.fill_(2.0)
    logits, _ = model(torch.randn(2, 32, 32, 3))
    assert torch.allclose(logits[:, :3], torch.ones(2, 3))
    assert torch.allclose(logits[:, 3:], 2 * torch.ones(2, 2))


def test_weight_align_math():
    model = CilModel("resnet20", 32)
    model.prev_model_adaption(4)
    model.prev_model_adaption(4)
    with torch.no_grad():
        model.fc.heads[0].weight.copy_(torch.eye(4, 64) * 2.0)  # row norms 2
        model.fc.heads[1].weight.copy_(torch.eye(4, 64) * 8.0)  # row norms 8
    gamma = model.weight_align(4)
    assert abs(gamma - 0.25) < 1e-6
    norms = model.fc.heads[1].weight.norm(dim=1)
    assert torch.allclose(norms, torch.full((4,), 2.0), atol=1e-5)


def test_weight_align_only_last_head():
    model = CilModel("resnet20", 32)
    model.prev_model_adaption(4)
    model.prev_model_adaption(4)
    w0 = model.fc.heads[0].weight.clone()
    model.weight_align(4)
    assert torch.equal(w0, model.fc.heads[0].weight)


def test_freeze_all():
    model = CilModel("resnet20", 32)
    model.prev_model_adaption(4)
    t = model.copy()
    t.freeze(["all"])
    assert all(not p.requires_grad for p in t.parameters())
    assert not t.training
    # original untouched
    assert all(p.requires_grad for p in model.parameters())


def test_after_model_adaption_skips_task0():
    model = CilModel("resnet20", 32)
    model.prev_model_adaption(4)
    w = model.fc.heads[0].weight.clone()

    class A:
        task_id = 0
    model.after_model_adaption(4, A())
    assert torch.equal(w, model.fc.heads[0].weight)
