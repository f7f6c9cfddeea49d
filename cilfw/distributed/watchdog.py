"""Failure detection — rank heartbeat + collective-timeout surfacing.

The reference has none: a dead rank hangs every other rank
forever at the next barrier/all-reduce (template.py:272, utils.py:39).
cilfw's failure story has two layers:

1. **Collective timeout**: the process group is created with a finite
   timeout (``--dist_timeout``, cilfw/distributed/init.py), so a collective
   stuck on a dead peer raises instead of hanging (gloo enforces this
   natively; for RCCL, init sets TORCH_NCCL_ASYNC_ERROR_HANDLING so the
   NCCL watchdog turns a stuck collective into an error). The engine wraps
   the task loop, prints the last completed checkpoint to resume from, and
   exits non-zero.
2. **Heartbeat watchdog** (this module): a daemon thread that checks the
   training loop is still making progress — it covers hangs that never
   reach a collective (a stuck data loader, a deadlocked host thread).
   The loop calls ``beat()`` every step; if no beat arrives within the
   timeout the watchdog prints a diagnostic with the resume pointer and
   hard-exits the process so an external launcher (torchrun) can tear the
   job down and restart from the checkpoint.
"""

import os
import sys
import threading
import time


EXIT_CODE = 87  # distinct code: "cilfw watchdog killed a stalled rank"


class Watchdog:
    def __init__(self, timeout_s, rank=0, checkpoint_dir=""):
        self.timeout_s = timeout_s
        self.rank = rank
        self.checkpoint_dir = checkpoint_dir
        self._last = time.monotonic()
        self._stop = threading.Event()
        self._thread = None
        self.last_checkpoint = None

    def start(self):
        if self.timeout_s and self.timeout_s > 0 and self._thread is None:
            self._thread = threading.Thread(target=self._watch, daemon=True)
            self._thread.start()
        return self

    def beat(self):
        self._last = time.monotonic()

    def note_checkpoint(self, path):
        self.last_checkpoint = path

    def stop(self):
        self._stop.set()

    def _watch(self):
        while not self._stop.wait(min(self.timeout_s / 4.0, 5.0)):
            idle = time.monotonic() - self._last
            if idle > self.timeout_s:
                msg = (f"[cilfw watchdog] rank {self.rank}: no training "
                       f"progress for {idle:.0f}s (> {self.timeout_s}s); "
                       f"assuming a hung/dead peer. ")
                msg += (f"Resume from {self.last_checkpoint}"
                        if self.last_checkpoint else
                        "No checkpoint written yet (use --output_dir to "
                        "enable per-task checkpoints).")
                print(msg, file=sys.stderr, flush=True)
                os._exit(EXIT_CODE)


def describe_failure(exc, last_checkpoint):
    """One-line operator guidance for a surfaced collective failure."""
    hint = (f"resume with --resume {last_checkpoint}" if last_checkpoint
            else "no checkpoint was written (run with --output_dir)")
    return (f"[cilfw] distributed failure detected: "
            f"{type(exc).__name__}: {exc} — a peer rank likely died or "
            f"stalled past --dist_timeout; {hint}")
