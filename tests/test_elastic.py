"""Elastic restart supervisor (tools/run_elastic.py): a failed attempt must
be relaunched from the newest task checkpoint and complete the protocol.

The reference cannot do this at all — it has neither checkpoints nor failure
detection, so a crash in task N restarts the whole run."""

import importlib.util
import os
import subprocess
import sys
import textwrap

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

_spec = importlib.util.spec_from_file_location(
    "run_elastic", os.path.join(REPO, "tools", "run_elastic.py"))
run_elastic = importlib.util.module_from_spec(_spec)
_spec.loader.exec_module(run_elastic)


def test_newest_checkpoint(tmp_path):
    assert run_elastic.newest_checkpoint(str(tmp_path)) is None
    assert run_elastic.newest_checkpoint(str(tmp_path / "missing")) is None
    for n in (0, 2, 10):
        (tmp_path / f"task_{n}.pth").write_bytes(b"x")
    (tmp_path / "task_bad.pth").write_bytes(b"x")
    (tmp_path / "other.pth").write_bytes(b"x")
    assert run_elastic.newest_checkpoint(str(tmp_path)).endswith("task_10.pth")


def test_strip_resume_and_build_cmd():
    base = ["python", "t.py", "--resume", "old.pth", "--lr", "0.1",
            "--resume=older.pth"]
    assert run_elastic.strip_resume(base) == ["python", "t.py", "--lr", "0.1"]
    cmd = run_elastic.build_cmd(base, nproc=0, resume="new.pth")
    assert cmd == ["python", "t.py", "--lr", "0.1", "--resume", "new.pth"]
    # torchrun wrapping: loopback rendezvous, script keeps its args
    cmd = run_elastic.build_cmd(["python", "t.py", "--lr", "0.1"],
                                nproc=2, resume=None)
    assert cmd[:4] == [sys.executable, "-m", "torch.distributed.run",
                       "--nnodes=1"]
    assert "--nproc-per-node=2" in cmd and "127.0.0.1" in cmd
    assert cmd[-3:] == ["t.py", "--lr", "0.1"]


def test_gives_up_without_progress(tmp_path):
    """A hard-failing child (never writes a checkpoint) must not loop
    forever: after max_restarts+1 attempts the child's code is returned."""
    p = subprocess.run(
        [sys.executable, os.path.join(REPO, "tools", "run_elastic.py"),
         "--output_dir", str(tmp_path), "--max_restarts", "1",
         "--backoff", "0.05", "--",
         sys.executable, "-c", "import sys; sys.exit(7)"],
        capture_output=True, text=True, timeout=120)
    assert p.returncode == 7, p.stdout + p.stderr
    assert p.stdout.count("attempt") >= 2
    assert "giving up" in p.stdout


@pytest.mark.timeout(600)
def test_crash_resume_completes_protocol(tmp_path):
    """End to end: the training process dies with the watchdog code (87) at
    the start of task 1 on the first attempt; the supervisor relaunches with
    --resume task_0.pth and the 2-task protocol completes with both acc1s."""
    outdir = tmp_path / "ckpts"
    sentinel = tmp_path / "crashed.once"
    child = tmp_path / "crashy.py"
    child.write_text(textwrap.dedent(f"""
        import os, sys
        sys.path.insert(0, {REPO!r})
        import cilfw.engine as E
        _orig = E.train_one_task
        def wrapped(*a, **k):
            # first attempt only: die right after task 0's checkpoint exists
            if (os.path.exists(os.path.join({str(outdir)!r}, "task_0.pth"))
                    and not os.path.exists({str(sentinel)!r})):
                open({str(sentinel)!r}, "w").close()
                os._exit(87)
            return _orig(*a, **k)
        E.train_one_task = wrapped
        from cilfw.config import parse_args
        args = parse_args([
            "--data_set", "synthetic", "--backbone", "resnet20",
            "--synthetic_classes", "10", "--num_bases", "5",
            "--increment", "5", "--num_epochs", "2", "--batch_size", "16",
            "--workers", "0", "--synthetic_train_size", "160",
            "--memory_size", "20", "--eval_every_epoch", "0",
            "--input_size", "16", "--no_aug", "--max_tasks", "2",
            "--output_dir", {str(outdir)!r},
        ] + sys.argv[1:])
        E.run(args)
    """))
    p = subprocess.run(
        [sys.executable, os.path.join(REPO, "tools", "run_elastic.py"),
         "--output_dir", str(outdir), "--max_restarts", "2",
         "--backoff", "0.05", "--", sys.executable, str(child)],
        capture_output=True, text=True, timeout=540, cwd=REPO)
    assert p.returncode == 0, p.stdout[-3000:] + p.stderr[-3000:]
    assert sentinel.exists(), "fault was never injected"
    assert "exited 87 (heartbeat watchdog" in p.stdout
    assert "attempt 2 (resume" in p.stdout
    assert (outdir / "task_1.pth").exists()
    # the resumed attempt carried task 0's accuracy through the checkpoint
    assert "task id = 1" in p.stdout
    last = [l for l in p.stdout.splitlines() if "acc1s" in l][-1]
    assert last.count(",") >= 1, f"expected 2 acc1s entries: {last}"


@pytest.mark.timeout(600)
def test_nproc_wrapping_runs_torchrun_gloo(tmp_path):
    """--nproc 2 wraps the child in torch.distributed.run with a loopback
    rendezvous and a fresh port; a tiny 1-task 2-rank gloo protocol must
    complete cleanly through the supervisor."""
    outdir = tmp_path / "ckpts"
    child = tmp_path / "train2.py"
    child.write_text(textwrap.dedent(f"""
        import sys
        sys.path.insert(0, {REPO!r})
        from cilfw.config import parse_args
        from cilfw.engine import run
        args = parse_args([
            "--data_set", "synthetic", "--backbone", "resnet20",
            "--synthetic_classes", "10", "--num_bases", "5",
            "--increment", "5", "--num_epochs", "1", "--batch_size", "16",
            "--workers", "0", "--synthetic_train_size", "128",
            "--memory_size", "20", "--eval_every_epoch", "0",
            "--input_size", "16", "--no_aug", "--max_tasks", "1",
            "--output_dir", {str(outdir)!r},
        ] + sys.argv[1:])
        run(args)
    """))
    p = subprocess.run(
        [sys.executable, os.path.join(REPO, "tools", "run_elastic.py"),
         "--output_dir", str(outdir), "--max_restarts", "0", "--nproc", "2",
         "--", sys.executable, str(child)],
        capture_output=True, text=True, timeout=540, cwd=REPO)
    assert p.returncode == 0, p.stdout[-3000:] + p.stderr[-3000:]
    assert "torch.distributed.run" in p.stdout
    assert (outdir / "task_0.pth").exists()
