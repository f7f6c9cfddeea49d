"""Lightweight tracing — rocTX-style ranges around task/epoch/step.

The reference has zero instrumentation. cilfw emits roctx ranges
through torch.cuda.nvtx (which maps to roctx on ROCm), so
``rocprofv3 --marker-trace`` shows the task/epoch/step structure around the
kernel trace. No-ops cleanly on CPU.
"""

from contextlib import contextmanager

import torch

_enabled = torch.cuda.is_available()


@contextmanager
def trace_range(name):
    if _enabled:
        try:
            torch.cuda.nvtx.range_push(name)
        except Exception:
            yield
            return
        try:
            yield
        finally:
            torch.cuda.nvtx.range_pop()
    else:
        yield
