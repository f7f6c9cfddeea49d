"""Class-incremental scenario engine — the continuum ClassIncremental contract.

Required behavior (consumed at reference template.py:223-231,
300-302):
- split a labeled dataset into T tasks by ``class_order``: task 0 = first
  ``initial_increment`` classes, then ``increment`` per task;
- labels are REMAPPED so position in class_order == label index;
- ``len(scenario)`` = T; ``scenario[t]`` yields a TaskSet of (image, label,
  task_id) triples; slicing ``scenario[:t+1]`` merges tasks 0..t (cumulative eval);
- ``scenario.nb_classes``;
- TaskSet.add_samples(x, y, t) appends replay exemplars;
- TaskSet.get_raw_samples() returns the untransformed (x, y, t) arrays.
"""

import numpy as np
import torch
from torch.utils.data import Dataset


class TaskSet(Dataset):
    """One task's samples (+ appended replay exemplars), with a transform."""

    def __init__(self, x, y, t, transform=None):
        self.x = x                      # uint8 (N,H,W,C)
        self.y = np.asarray(y, dtype=np.int64)
        self.t = np.asarray(t, dtype=np.int64)
        self.transform = transform

    def __len__(self):
        return len(self.y)

    def __getitem__(self, i):
        img = self.x[i]
        if isinstance(img, (str, bytes)) or (isinstance(img, np.str_)):
            from .datasets import load_image
            img = load_image(img)
        if self.transform is not None:
            img = self.transform(img)
        else:
            img = torch.from_numpy(
                np.ascontiguousarray(img).copy()).float().div_(255.0)
        return img, int(self.y[i]), int(self.t[i])

    def add_samples(self, x, y, t):
        self.x = np.concatenate([self.x, x])
        self.y = np.concatenate([self.y, np.asarray(y, dtype=np.int64)])
        self.t = np.concatenate([self.t, np.asarray(t, dtype=np.int64)])

    def get_raw_samples(self):
        return self.x, self.y, self.t

    @property
    def nb_classes(self):
        return len(np.unique(self.y))


class ClassIncremental:
    """Splits (x, y) into tasks of [initial_increment, increment, increment, ...]."""

    def __init__(self, x, y, initial_increment, increment, class_order=None,
                 transform=None):
        y = np.asarray(y, dtype=np.int64)
        classes = np.unique(y)
        self.nb_classes = len(classes)
        if class_order is None:
            class_order = list(range(self.nb_classes))
        assert len(class_order) == self.nb_classes
        self.class_order = list(class_order)
        if initial_increment == 0:
            initial_increment = increment
        self.initial_increment = initial_increment
        self.increment = increment
        rest = self.nb_classes - initial_increment
        assert rest >= 0 and rest % increment == 0, \
            f"{self.nb_classes} classes don't split into {initial_increment} + " \
            f"k*{increment}"
        self.nb_tasks = 1 + rest // increment

        # label remap: original class id -> position in class_order
        remap = np.empty(int(classes.max()) + 1, dtype=np.int64)
        remap[np.asarray(self.class_order)] = np.arange(self.nb_classes)
        self._x = x
        self._y_remapped = remap[y]
        self.transform = transform

        # per-task class ranges in remapped space
        self.task_bounds = []
        start = 0
        for t in range(self.nb_tasks):
            size = initial_increment if t == 0 else increment
            self.task_bounds.append((start, start + size))
            start += size

    def __len__(self):
        return self.nb_tasks

    def increments(self, t):
        lo, hi = self.task_bounds[t]
        return hi - lo

    def _gather(self, tasks):
        xs, ys, ts = [], [], []
        for t in tasks:
            lo, hi = self.task_bounds[t]
            mask = (self._y_remapped >= lo) & (self._y_remapped < hi)
            xs.append(self._x[mask])
            ys.append(self._y_remapped[mask])
            ts.append(np.full(mask.sum(), t, dtype=np.int64))
        return (np.concatenate(xs), np.concatenate(ys), np.concatenate(ts))

    def __getitem__(self, key):
        if isinstance(key, slice):
            tasks = range(*key.indices(self.nb_tasks))
            x, y, t = self._gather(tasks)
            return TaskSet(x, y, t, self.transform)
        if key < 0:
            key += self.nb_tasks
        x, y, t = self._gather([key])
        return TaskSet(x, y, t, self.transform)
