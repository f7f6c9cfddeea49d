"""GPU end-to-end: a tiny 2-task WA run on the HIP path (bf16) must learn,
be deterministic, and keep the CIL semantics intact (pytest -m gpu)."""

import pytest
import torch

from cilfw.config import parse_args
from cilfw.engine import run

pytestmark = [pytest.mark.gpu,
              pytest.mark.skipif(not torch.cuda.is_available(),
                                 reason="needs GPU")]


def _args(epochs=4):
    return parse_args([
        "--data_set", "synthetic", "--backbone", "resnet18",
        "--synthetic_classes", "20",
        "--num_bases", "15", "--increment", "5",
        "--num_epochs", str(epochs), "--batch_size", "64", "--workers", "0",
        "--synthetic_train_size", "1600", "--memory_size", "60",
        "--eval_every_epoch", "0", "--input_size", "32", "--no_aug",
        "--lr", "0.05", "--seed", "3", "--dtype", "bf16",
    ])


@pytest.mark.timeout(600)
def test_gpu_two_task_run_learns():
    accs = run(_args())
    assert len(accs) == 2
    assert accs[0] > 40.0, f"GPU base-task accuracy too low: {accs}"
    assert all(torch.isfinite(torch.tensor(a)) for a in accs)


@pytest.mark.timeout(600)
def test_gpu_data_pipeline_run():
    """HBM-resident data path (--gpu_data) end-to-end."""
    args = _args(epochs=3)
    args.gpu_data = True
    accs = run(args)
    assert len(accs) == 2
    assert accs[0] > 40.0, f"--gpu_data run failed to learn: {accs}"


@pytest.mark.timeout(600)
def test_gpu_resnet50_step():
    """Bottleneck blocks + 7x7 stem + maxpool on the HIP path (ImageNet-style
    geometry at reduced size)."""
    from cilfw.models import CilModel
    from cilfw import ops
    from cilfw.distributed import DataParallelEngine
    from cilfw.optim import FlatSGD
    torch.manual_seed(0)
    model = CilModel("resnet50", 224).to("cuda")  # 7x7/s2 stem + maxpool
    model.prev_model_adaption(10)
    model = model.to("cuda")
    engine = DataParallelEngine(model)
    opt = FlatSGD(engine, lr=0.1)
    x = torch.randn(8, 64, 64, 3, device="cuda").to(torch.bfloat16)
    y = torch.randint(0, 10, (8,), device="cuda")
    for _ in range(2):
        opt.zero_grad()
        logits, feats = model(x)
        loss = ops.cross_entropy(logits.float(), y)
        loss.backward()
        engine.finalize()
        opt.step()
    torch.cuda.synchronize()
    assert feats.shape == (8, 2048)
    assert torch.isfinite(loss)


@pytest.mark.timeout(600)
def test_gpu_determinism():
    """No fp32 atomics anywhere in the step: two identical runs bit-match."""
    a1 = run(_args(epochs=2))
    a2 = run(_args(epochs=2))
    assert a1 == a2, f"GPU training is not deterministic: {a1} vs {a2}"
