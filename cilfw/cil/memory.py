"""Rehearsal memory + herding exemplar selection (continuum RehearsalMemory parity).

Contract (reference template.py:212-216, 231, 300-302):
- capacity-bounded store (memory_size total); per-class quota =
  memory_size // nb_seen_classes (shrinks as classes arrive), or
  memory_size // total_nb_classes when fixed_memory;
- add(x, y, t, features): herding 'barycenter' = iCaRL greedy — per class,
  iteratively pick the candidate minimizing || class-feature-mean -
  mean(selected + candidate) || (README.md:136 of the reference, O(n*m));
- get() -> all stored (x, y, t).

The selection itself runs on whatever device the features live on: pass GPU
features and the greedy argmin runs on-device (HIP herding kernel for the
candidate-distance scan; torch ops on CPU).
"""

import numpy as np
import torch

from ..ops._backend import use_hip, ext


def herding_select(features, m):
    """Greedy barycenter herding: return indices (ranked) of <=m exemplars.

    features: (n, D) float tensor (any device). Deterministic; ties break to the
    lowest index (torch.argmin semantics), replicated identically on every rank.
    """
    f = features.float()
    n = f.shape[0]
    m = min(m, n)
    mu = f.mean(dim=0)
    if f.is_cuda and use_hip(f):
        return ext().herding_select(f.contiguous(), mu.contiguous(), m)
    selected = torch.zeros(n, dtype=torch.bool, device=f.device)
    sum_sel = torch.zeros_like(mu)
    order = torch.empty(m, dtype=torch.int64, device=f.device)
    for k in range(m):
        cand_mean = (sum_sel.unsqueeze(0) + f) / (k + 1)
        d = (cand_mean - mu).pow(2).sum(dim=1)
        d[selected] = float("inf")
        i = torch.argmin(d)
        order[k] = i
        selected[i] = True
        sum_sel += f[i]
    return order


class RehearsalMemory:
    def __init__(self, memory_size=2000, herding_method="barycenter",
                 fixed_memory=False, nb_total_classes=None):
        self.memory_size = memory_size
        self.herding_method = herding_method
        self.fixed_memory = fixed_memory
        self.nb_total_classes = nb_total_classes
        if fixed_memory:
            assert nb_total_classes, "fixed_memory needs nb_total_classes"
        # per-class stores, in insertion-ranked (herding) order
        self._x = {}   # class -> uint8 (k,H,W,C)
        self._y = {}
        self._t = {}
        # last add()'s selection: class -> global indices (into the x passed
        # to add, ranked) — lets a device-resident mirror gather the same
        # exemplars from an already-uploaded task tensor with no host traffic
        self.last_selection = {}
        self.last_quota = None

    @property
    def nb_classes(self):
        return len(self._x)

    def _quota(self, nb_seen):
        if self.fixed_memory:
            return self.memory_size // self.nb_total_classes
        return self.memory_size // max(nb_seen, 1)

    def add(self, x, y, t, features):
        """x,y,t: raw task arrays (incl. replayed old-class samples — filtered);
        features: (len(x), D) tensor aligned with x."""
        y = np.asarray(y)
        t = np.asarray(t)
        if isinstance(features, np.ndarray):
            features = torch.from_numpy(features)
        new_classes = [c for c in np.unique(y) if c not in self._x]
        nb_seen = self.nb_classes + len(new_classes)
        quota = self._quota(nb_seen)
        self.last_selection = {}
        self.last_quota = quota
        idxs = {c: np.where(y == c)[0] for c in new_classes}
        ranked_by_class = {}
        if self.herding_method == "barycenter" and features.is_cuda \
                and len(new_classes) > 1:
            # all classes in ONE launch — concurrent per-class greedy loops
            from ..ops._backend import use_hip, ext
            if use_hip(features):
                feats = [features[idxs[c]] for c in new_classes]
                # rank only the kept prefix (greedy prefix property: the
                # first q picks are identical whatever the total m)
                orders = ext().herding_select_batch(
                    feats, [quota for _ in new_classes])
                for c, o in zip(new_classes, orders):
                    ranked_by_class[c] = o.cpu().numpy()
        for c in new_classes:
            idx = idxs[c]
            if c in ranked_by_class:
                ranked = ranked_by_class[c]
            elif self.herding_method == "barycenter":
                ranked = herding_select(features[idx], quota).cpu().numpy()
            elif self.herding_method == "random":
                rng = np.random.default_rng(int(c))
                ranked = rng.permutation(len(idx))
            else:
                raise ValueError(self.herding_method)
            keep = idx[ranked[:quota]]
            self.last_selection[int(c)] = keep
            self._x[int(c)] = np.ascontiguousarray(x[keep])
            self._y[int(c)] = y[keep]
            self._t[int(c)] = t[keep]
        # shrink old classes to the new quota (herding rank order is kept)
        for c in list(self._x):
            self._x[c] = self._x[c][:quota]
            self._y[c] = self._y[c][:quota]
            self._t[c] = self._t[c][:quota]

    def get(self):
        classes = sorted(self._x)
        if not classes:
            raise ValueError("memory is empty")
        return (np.concatenate([self._x[c] for c in classes]),
                np.concatenate([self._y[c] for c in classes]),
                np.concatenate([self._t[c] for c in classes]))

    def __len__(self):
        return sum(len(v) for v in self._y.values())

    def state_dict(self):
        return {"memory_size": self.memory_size, "herding": self.herding_method,
                "fixed": self.fixed_memory, "total": self.nb_total_classes,
                "x": self._x, "y": self._y, "t": self._t}

    def load_state_dict(self, sd):
        self.memory_size = sd["memory_size"]
        self.herding_method = sd["herding"]
        self.fixed_memory = sd["fixed"]
        self.nb_total_classes = sd["total"]
        self._x, self._y, self._t = sd["x"], sd["y"], sd["t"]
