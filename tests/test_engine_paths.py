"""Engine feature-path coverage: dynamic lambda_kd, fixed memory, periodic
eval, CLI parse of the reference flag surface."""

import subprocess
import sys

import pytest

from cilfw.config import parse_args
from cilfw.engine import run


def _args(extra=(), epochs=2):
    return parse_args([
        "--data_set", "synthetic", "--backbone", "resnet20",
        "--synthetic_classes", "20", "--num_bases", "10", "--increment", "5",
        "--num_epochs", str(epochs), "--batch_size", "32", "--workers", "0",
        "--synthetic_train_size", "800", "--memory_size", "40",
        "--eval_every_epoch", "0", "--input_size", "16", "--no_aug",
        "--lr", "0.05", "--seed", "1",
    ] + list(extra))


@pytest.mark.timeout(900)
def test_dynamic_lambda_and_fixed_memory_and_eval_every():
    args = _args(["--dynamic_lambda_kd", "--fixed_memory",
                  "--eval_every_epoch", "1", "--herding_method", "random"])
    accs = run(args)
    assert len(accs) == 3  # 10 + 5 + 5


def test_reference_cli_surface():
    """All 28 reference flags parse."""
    argv = ["--seed", "7", "--num_bases", "50", "--increment", "10",
            "--backbone", "resnet32", "--batch_size", "128",
            "--input_size", "32", "--color_jitter", "0.4",
            "--aa", "rand-m9-mstd0.5-inc1", "--train_interpolation", "bicubic",
            "--reprob", "0.25", "--remode", "pixel", "--recount", "1",
            "--herding_method", "barycenter", "--memory_size", "2000",
            "--fixed_memory", "--lr", "0.1", "--momentum", "0.9",
            "--weight_decay", "5e-4", "--num_epochs", "140",
            "--smooth", "0.1", "--eval_every_epoch", "5",
            "--dist_url", "env://", "--data_set", "cifar100",
            "--data_path", "/data/cifar100", "--lambda_kd", "0.5",
            "--dynamic_lambda_kd", "--resplit"]
    args = parse_args(argv)
    assert args.num_bases == 50 and args.memory_size == 2000
    assert args.lambda_kd == 0.5 and args.dynamic_lambda_kd


def test_template_entrypoint_help():
    out = subprocess.run([sys.executable, "template.py", "--help"],
                         capture_output=True, text=True, timeout=120)
    assert out.returncode == 0
    assert "--num_bases" in out.stdout


@pytest.mark.timeout(900)
def test_imagenet_config_paths_and_metric_every():
    """BASELINE configs 3/5 plumbing (synthetic stand-ins) + sampled metrics."""
    args = parse_args([
        "--data_set", "imagenet100", "--backbone", "resnet18",
        "--num_bases", "50", "--increment", "50", "--num_epochs", "1",
        "--batch_size", "16", "--workers", "0", "--input_size", "32",
        "--no_aug", "--memory_size", "40", "--eval_every_epoch", "0",
        "--metric_every", "4",
    ])
    accs = run(args)
    assert len(accs) == 2


@pytest.mark.timeout(900)
def test_baseline_config0_b0inc10_resnet32():
    """BASELINE.json config[0]: CIFAR-100-shaped B0-Inc10 ResNet-32 on CPU,
    world_size=1, first 2 tasks only."""
    args = parse_args([
        "--data_set", "synthetic", "--backbone", "resnet32",
        "--num_bases", "0", "--increment", "10", "--max_tasks", "2",
        "--num_epochs", "1", "--batch_size", "32", "--workers", "0",
        "--synthetic_train_size", "2000", "--memory_size", "200",
        "--eval_every_epoch", "0", "--input_size", "32", "--no_aug",
    ])
    accs = run(args)
    assert len(accs) == 2
    assert args.known_classes == 20
