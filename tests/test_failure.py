"""Failure detection: a dead rank must surface as a timely error with a
resume pointer, not an indefinite hang (the reference hangs at the next
barrier — template.py:272; SURVEY.md §5 flags the absence)."""

import os
import time

import pytest
import torch.multiprocessing as mp

from cilfw.distributed.watchdog import Watchdog, describe_failure


def _worker(rank, world, port, out):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["LOCAL_RANK"] = str(rank)
    from cilfw.config import parse_args
    from cilfw.engine import run
    args = parse_args([
        "--data_set", "synthetic", "--backbone", "resnet20",
        "--synthetic_classes", "10", "--num_bases", "5", "--increment", "5",
        "--num_epochs", "2", "--batch_size", "16", "--workers", "0",
        "--synthetic_train_size", "160", "--memory_size", "20",
        "--eval_every_epoch", "0", "--input_size", "16", "--no_aug",
        "--max_tasks", "1", "--dist_timeout", "5",
    ])
    if rank == 1:
        import cilfw.engine as E

        def die(*a, **k):  # simulated rank death before training collectives
            os._exit(0)
        E.train_one_task = die
    t0 = time.time()
    try:
        run(args)
        code = 0
    except SystemExit as e:
        code = e.code
    out[rank] = (code, time.time() - t0)
    os._exit(0)  # skip destroy_process_group (peer is gone)


@pytest.mark.timeout(240)
def test_dead_rank_surfaces_within_timeout():
    """Rank 1 dies mid-run; rank 0 must exit with the failure code within
    the collective timeout instead of hanging."""
    mgr = mp.Manager()
    out = mgr.dict()
    mp.spawn(_worker, args=(2, 29655, out), nprocs=2, join=True)
    code, dur = out[0]
    assert code == 3, f"survivor must exit via the failure path, got {code}"
    assert dur < 120, f"failure took {dur:.0f}s — timeout not effective"


def test_watchdog_beats_and_fires(tmp_path):
    """Watchdog fires only when beats stop; runs the firing path in a
    subprocess (it hard-exits)."""
    import subprocess
    import sys
    wd = Watchdog(timeout_s=2.0, rank=0).start()
    for _ in range(3):
        wd.beat()
        time.sleep(0.2)
    assert wd._thread.is_alive()
    wd.stop()

    code = (
        "import time\n"
        "from cilfw.distributed.watchdog import Watchdog, EXIT_CODE\n"
        "wd = Watchdog(timeout_s=1.0, rank=0).start()\n"
        "wd.note_checkpoint('ckpt/task_3.pth')\n"
        "time.sleep(30)\n"
    )
    t0 = time.time()
    p = subprocess.run([sys.executable, "-c", code], capture_output=True,
                       text=True, timeout=60,
                       cwd=os.path.dirname(os.path.dirname(__file__)))
    assert p.returncode == 87, p.stderr
    assert "task_3.pth" in p.stderr
    assert time.time() - t0 < 30


def test_describe_failure_mentions_resume():
    msg = describe_failure(RuntimeError("Timed out"), "out/task_2.pth")
    assert "task_2.pth" in msg and "Timed out" in msg
    msg2 = describe_failure(RuntimeError("x"), None)
    assert "no checkpoint" in msg2
