"""Reference-exact artifacts: the hardcoded CIFAR-100 class order and the
printed run schema people diff against reference transcripts.

The class-order list is the reference's exact constant (template.py:201-202
— a data constant, copied deliberately so per-task accuracy trajectories are
directly comparable across frameworks). The schema test runs a seeded
2-task protocol and checks each load-bearing printed line against the
reference's formats (template.py:186, :289; engine.py mirrors them)."""

import io
import re
import contextlib

import pytest

from cilfw.config import parse_args
from cilfw.data import CIFAR100_CLASS_ORDER
from cilfw.engine import run

# verbatim from reference template.py:201-202
REFERENCE_ORDER = [
    68, 56, 78, 8, 23, 84, 90, 65, 74, 76, 40, 89, 3, 92, 55, 9, 26, 80, 43,
    38, 58, 70, 77, 1, 85, 19, 17, 50, 28, 53, 13, 81, 45, 82, 6, 59, 83, 16,
    15, 44, 91, 41, 72, 60, 79, 52, 20, 10, 31, 54, 37, 95, 14, 71, 96, 98,
    97, 2, 64, 66, 42, 22, 35, 86, 24, 34, 87, 21, 99, 0, 88, 27, 18, 94, 11,
    12, 47, 25, 30, 46, 62, 69, 36, 61, 7, 63, 75, 5, 32, 4, 51, 48, 73, 93,
    39, 67, 29, 49, 57, 33,
]


def test_class_order_is_reference_exact():
    assert CIFAR100_CLASS_ORDER == REFERENCE_ORDER
    assert sorted(CIFAR100_CLASS_ORDER) == list(range(100))


@pytest.mark.timeout(900)
def test_printed_schema_matches_reference_formats():
    args = parse_args([
        "--data_set", "synthetic", "--backbone", "resnet20",
        "--synthetic_classes", "10", "--num_bases", "5", "--increment", "5",
        "--num_epochs", "2", "--batch_size", "32", "--workers", "0",
        "--synthetic_train_size", "320", "--memory_size", "40",
        "--eval_every_epoch", "2", "--input_size", "16", "--no_aug",
        "--lr", "0.05", "--seed", "3", "--max_tasks", "2",
    ])
    buf = io.StringIO()
    with contextlib.redirect_stdout(buf):
        accs = run(args)
    out = buf.getvalue()

    # per-task line — reference template.py:289 format, diffable verbatim
    task_lines = re.findall(
        r"^task id = (\d+)  @Acc1 = (\d+\.\d{5}), acc1s = \[.*\]$",
        out, re.M)
    assert len(task_lines) == 2
    assert float(task_lines[0][1]) == pytest.approx(accs[0], abs=1e-4)

    # eval line — reference template.py:186 shape (Acc@1 ... loss ...)
    assert re.search(r"\* Acc@1 \d+\.\d{3} .*loss \d+\.\d{3}", out)

    # weight-align gamma report (reference template.py:165 prints the
    # norms; cilfw prints old/new norm and gamma)
    assert re.search(r"old norm: \d+\.\d+, new norm: \d+\.\d+, "
                     r"gamma: \d+\.\d+", out)

    # per-epoch train meters include ce/kd/loss/acc1 and lr
    epoch_lines = [ln for ln in out.splitlines()
                   if re.match(r"^task \d+ epoch \d+:", ln)]
    assert len(epoch_lines) == 4  # 2 tasks x 2 epochs
    for ln in epoch_lines:
        for key in ("lr:", "ce:", "kd:", "loss:", "acc1:", "imgs/s"):
            assert key in ln, (key, ln)

    # final summary
    assert re.search(r"^average incremental accuracy = \d+\.\d{5}$", out,
                     re.M)
