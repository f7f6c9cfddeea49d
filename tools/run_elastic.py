#!/usr/bin/env python3
"""Elastic restart supervisor — closes the failure-detection loop.

The reference has no failure story at all (a dead rank hangs
every peer forever at the next collective, /root/reference/template.py:272).
cilfw detects the failure (finite collective timeouts + heartbeat watchdog,
cilfw/distributed/watchdog.py) and exits with a resume pointer; this
supervisor is the piece that ACTS on it: it launches the training command,
watches the exit code, and on failure relaunches from the newest per-task
checkpoint in --output_dir.

    python tools/run_elastic.py --output_dir ckpts --max_restarts 3 \
        [--nproc 8] -- python template.py --output_dir ckpts ...

Behavior:
  * exit 0            -> done, supervisor exits 0.
  * any nonzero exit  -> relaunch with ``--resume <newest task_*.pth>``
    (87 = cilfw watchdog killed a stalled rank; 3 = surfaced collective
    failure; anything else = crash). Any ``--resume`` already present in the
    child command is replaced.
  * restart budget is PROGRESS-BASED: --max_restarts bounds consecutive
    failures with no new checkpoint; a task boundary reached resets the
    count, so a long job survives many transient faults but a hard fault
    loop still terminates.
  * --nproc N wraps the command in ``python -m torch.distributed.run
    --nnodes=1 --nproc-per-node N --master-addr 127.0.0.1`` with a fresh
    rendezvous port per attempt (a restart must not collide with a
    lingering TIME_WAIT port).

The child runs in its own process group; on SIGINT/SIGTERM the supervisor
kills exactly that group (never a pattern match) and exits.
"""

import argparse
import os
import re
import signal
import socket
import subprocess
import sys
import time


def newest_checkpoint(output_dir):
    """Highest task_N.pth in output_dir, or None."""
    if not output_dir or not os.path.isdir(output_dir):
        return None
    best, best_n = None, -1
    for name in os.listdir(output_dir):
        m = re.fullmatch(r"task_(\d+)\.pth", name)
        if m and int(m.group(1)) > best_n:
            best_n = int(m.group(1))
            best = os.path.join(output_dir, name)
    return best


def strip_resume(cmd):
    out, skip = [], False
    for tok in cmd:
        if skip:
            skip = False
            continue
        if tok == "--resume":
            skip = True
            continue
        if tok.startswith("--resume="):
            continue
        out.append(tok)
    return out


def free_port():
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def build_cmd(base, nproc, resume):
    cmd = strip_resume(list(base))
    if resume:
        cmd += ["--resume", resume]
    if nproc and nproc > 1:
        # base is ["python", "script.py", ...] -> torchrun the script
        script = cmd[1:] if os.path.basename(cmd[0]).startswith("python") else cmd
        cmd = [sys.executable, "-m", "torch.distributed.run",
               "--nnodes=1", f"--nproc-per-node={nproc}",
               "--master-addr", "127.0.0.1",
               "--master-port", str(free_port())] + script
    return cmd


def main(argv=None):
    p = argparse.ArgumentParser(
        description="relaunch training from the last task checkpoint on failure")
    p.add_argument("--output_dir", required=True,
                   help="checkpoint dir (must match the child's --output_dir)")
    p.add_argument("--max_restarts", type=int, default=3,
                   help="max consecutive failures WITHOUT a new checkpoint")
    p.add_argument("--nproc", type=int, default=0,
                   help="wrap the command in torch.distributed.run with N ranks")
    p.add_argument("--backoff", type=float, default=1.0,
                   help="seconds to wait before a relaunch")
    p.add_argument("cmd", nargs=argparse.REMAINDER,
                   help="-- <training command>")
    args = p.parse_args(argv)

    base = args.cmd[1:] if args.cmd[:1] == ["--"] else args.cmd
    if not base:
        p.error("no training command given (append: -- python template.py ...)")

    attempt, failures_since_progress = 0, 0
    child = None

    def forward_signal(signum, _frame):
        if child is not None and child.poll() is None:
            os.killpg(child.pid, signum)  # exact pgid we created, never a pattern
        raise SystemExit(128 + signum)

    signal.signal(signal.SIGINT, forward_signal)
    signal.signal(signal.SIGTERM, forward_signal)

    while True:
        resume = newest_checkpoint(args.output_dir)
        cmd = build_cmd(base, args.nproc, resume)
        attempt += 1
        print(f"[run_elastic] attempt {attempt}"
              + (f" (resume {resume})" if resume else " (fresh start)")
              + f": {' '.join(cmd)}", flush=True)
        child = subprocess.Popen(cmd, start_new_session=True)
        rc = child.wait()
        if rc == 0:
            print(f"[run_elastic] training finished after {attempt} attempt(s), "
                  f"{attempt - 1} restart(s)", flush=True)
            return 0
        after = newest_checkpoint(args.output_dir)
        progressed = after is not None and after != resume
        failures_since_progress = 0 if progressed else failures_since_progress + 1
        why = {87: "heartbeat watchdog (stalled rank)",
               3: "surfaced collective failure"}.get(rc, "crash")
        print(f"[run_elastic] attempt {attempt} exited {rc} ({why}); "
              f"newest checkpoint: {after}; "
              f"failures since last progress: {failures_since_progress}",
              flush=True)
        if failures_since_progress > args.max_restarts:
            print(f"[run_elastic] no progress in {args.max_restarts + 1} "
                  f"consecutive attempts — giving up with exit {rc}", flush=True)
            return rc
        time.sleep(args.backoff)


if __name__ == "__main__":
    sys.exit(main())
