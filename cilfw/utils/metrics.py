"""Metrics & logging — SmoothedValue / MetricLogger capability parity.

Reference utils.py:22-118: windowed meters with exact cross-rank global averages via
an fp64 [count, total] all-reduce. cilfw batches ALL meters into ONE all-reduce per
sync (the reference did one collective per meter — SURVEY.md §2.3 N4).
"""

import datetime
import time
from collections import defaultdict, deque

import torch
import torch.distributed as dist


def is_dist():
    return dist.is_available() and dist.is_initialized()


class SmoothedValue:
    def __init__(self, window_size=20, fmt="{median:.4f} ({global_avg:.4f})"):
        self.deque = deque(maxlen=window_size)
        self.total = 0.0
        self.count = 0
        self.fmt = fmt

    def update(self, value, n=1):
        self.deque.append(value)
        self.count += n
        self.total += value * n

    @property
    def median(self):
        return torch.tensor(list(self.deque)).median().item() if self.deque else 0.0

    @property
    def avg(self):
        return torch.tensor(list(self.deque), dtype=torch.float64).mean().item() \
            if self.deque else 0.0

    @property
    def global_avg(self):
        return self.total / max(self.count, 1)

    @property
    def max(self):
        return max(self.deque) if self.deque else 0.0

    @property
    def value(self):
        return self.deque[-1] if self.deque else 0.0

    def __str__(self):
        return self.fmt.format(median=self.median, avg=self.avg,
                               global_avg=self.global_avg, max=self.max,
                               value=self.value)


class MetricLogger:
    def __init__(self, delimiter="  "):
        self.meters = defaultdict(SmoothedValue)
        self.delimiter = delimiter

    def update(self, **kwargs):
        for k, v in kwargs.items():
            if isinstance(v, torch.Tensor):
                v = v.item()
            self.meters[k].update(float(v))

    def update_n(self, n=1, **kwargs):
        for k, v in kwargs.items():
            if isinstance(v, torch.Tensor):
                v = v.item()
            self.meters[k].update(float(v), n=n)

    def __getattr__(self, attr):
        if attr in self.meters:
            return self.meters[attr]
        raise AttributeError(attr)

    def synchronize_between_processes(self, device=None):
        """One fp64 all-reduce for every meter's (count, total)."""
        if not is_dist():
            return
        names = sorted(self.meters.keys())
        if device is None:
            device = (torch.device("cuda") if torch.cuda.is_available()
                      else torch.device("cpu"))
        buf = torch.empty(2 * len(names), dtype=torch.float64, device=device)
        for i, n in enumerate(names):
            buf[2 * i] = self.meters[n].count
            buf[2 * i + 1] = self.meters[n].total
        dist.barrier()
        dist.all_reduce(buf)
        vals = buf.tolist()
        for i, n in enumerate(names):
            self.meters[n].count = int(vals[2 * i])
            self.meters[n].total = vals[2 * i + 1]

    def __str__(self):
        return self.delimiter.join(
            f"{name}: {meter}" for name, meter in self.meters.items())

    def log_every(self, iterable, print_freq, header=""):
        i = 0
        start = time.time()
        iter_time = SmoothedValue(fmt="{avg:.4f}")
        n_total = len(iterable)
        end = time.time()
        for obj in iterable:
            yield obj
            iter_time.update(time.time() - end)
            if print_freq and (i % print_freq == 0 or i == n_total - 1):
                eta = iter_time.avg * (n_total - i - 1)
                print(f"{header} [{i}/{n_total}] eta: "
                      f"{datetime.timedelta(seconds=int(eta))} {self} "
                      f"time: {iter_time}")
            i += 1
            end = time.time()
        total = time.time() - start
        if print_freq:
            print(f"{header} Total time: {datetime.timedelta(seconds=int(total))} "
                  f"({total / max(n_total, 1):.4f} s / it)")
