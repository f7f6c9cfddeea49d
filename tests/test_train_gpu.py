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
def test_gpu_determinism():
    """No fp32 atomics anywhere in the step: two identical runs bit-match."""
    a1 = run(_args(epochs=2))
    a2 = run(_args(epochs=2))
    assert a1 == a2, f"GPU training is not deterministic: {a1} vs {a2}"
