"""End-to-end CIL plumbing (BASELINE.json config[0]: CPU, world_size=1, 2 tasks).

Uses the class-separable synthetic dataset so a short run must actually learn;
checks the acc1s trajectory shape, WA invocation, determinism and resume."""

import pytest

from cilfw.config import parse_args
from cilfw.engine import run


def _args(extra=(), epochs=3):
    return parse_args([
        "--data_set", "synthetic", "--backbone", "resnet20",
        "--synthetic_classes", "20",
        "--num_bases", "15", "--increment", "5",
        "--num_epochs", str(epochs), "--batch_size", "32", "--workers", "0",
        "--synthetic_train_size", "1600", "--memory_size", "60",
        "--eval_every_epoch", "0", "--input_size", "16", "--no_aug",
        "--lr", "0.05", "--seed", "3",
    ] + list(extra))


@pytest.mark.timeout(900)
def test_two_task_run_learns_and_reports():
    args = _args(epochs=5)
    accs = run(args)
    assert len(accs) == 2
    # class-separable synthetic data: base task must be well above chance (5%)
    assert accs[0] > 40.0, f"base-task accuracy too low: {accs}"
    assert args.known_classes == 20


@pytest.mark.timeout(900)
def test_determinism_same_seed():
    a1 = run(_args(epochs=2))
    a2 = run(_args(epochs=2))
    assert a1 == a2


@pytest.mark.timeout(900)
def test_device_replay_matches_host_replay(monkeypatch):
    """Engine path: the HBM-resident DeviceReplayMirror replay source must
    produce the same trajectory as the host add_samples path (identical index
    space + shuffle + augmentation stream), here exercised on CPU via the
    CILFW_GPU_DATA_ON_CPU override."""
    monkeypatch.setenv("CILFW_GPU_DATA_ON_CPU", "1")
    host = run(_args(["--gpu_data", "--no_device_replay"], epochs=2))
    dev = run(_args(["--gpu_data"], epochs=2))
    assert host == dev


@pytest.mark.timeout(900)
def test_checkpoint_resume_rebuilds_device_mirror(tmp_path, monkeypatch):
    """Resume under --gpu_data: the HBM-resident replay mirror does not live
    in the checkpoint, so the engine rebuilds it from the restored host
    memory on the first post-resume task (engine.py "resumed run: one
    rebuild upload"). The resumed trajectory must match the uninterrupted
    run exactly."""
    monkeypatch.setenv("CILFW_GPU_DATA_ON_CPU", "1")
    base = ["--gpu_data", "--output_dir", str(tmp_path)]
    full = run(_args(base, epochs=2))
    resumed = run(_args(base + ["--resume", str(tmp_path / "task_0.pth")],
                        epochs=2))
    assert len(resumed) == 2
    assert resumed == full


@pytest.mark.timeout(900)
def test_checkpoint_resume_continues(tmp_path):
    base = ["--output_dir", str(tmp_path)]
    full = run(_args(base, epochs=2))
    resumed = run(_args(base + ["--resume", str(tmp_path / "task_0.pth")],
                        epochs=2))
    # resumed run re-does task 1 only; its task-0 acc comes from the checkpoint
    assert len(resumed) == 2
    assert resumed[0] == full[0]
