"""Walker/Vose alias tables for O(1) categorical sampling.

Replaces the reference's per-draw alias sampler (``random/AliasSampler.scala``,
``random/DiscreteDist.scala``). In this framework alias tables are built ONCE
(vectorised, numpy) for the *static* distributions — the empirical domain
distribution phi_a and the cached power distributions p_k(v) ∝ phi(v)*norm(v)^k
— and uploaded to the GPU as flat arrays. Dynamic (per-draw) categorical
sampling on the GPU uses Gumbel-max instead (no table build needed).
"""

from __future__ import annotations

import numpy as np


class AliasTable:
    """Vose alias table. ``prob`` and ``alias`` have length n.

    Draw: pick slot i uniformly, accept i with prob ``prob[i]`` else take
    ``alias[i]``.
    """

    __slots__ = ("prob", "alias", "probs")

    def __init__(self, weights: np.ndarray):
        w = np.asarray(weights, dtype=np.float64)
        if w.ndim != 1 or w.size == 0:
            raise ValueError("weights must be a non-empty 1-D array")
        if not np.all(np.isfinite(w)) or np.any(w < 0):
            raise ValueError("invalid weight encountered")
        total = w.sum()
        if total <= 0.0 or not np.isfinite(total):
            raise ValueError("zero or non-finite probability mass")
        p = w / total
        self.probs = p
        n = p.size
        scaled = p * n
        prob = np.empty(n, dtype=np.float64)
        alias = np.zeros(n, dtype=np.int64)
        small = [i for i in range(n) if scaled[i] < 1.0]
        large = [i for i in range(n) if scaled[i] >= 1.0]
        scaled = scaled.copy()
        while small and large:
            s = small.pop()
            l = large.pop()
            prob[s] = scaled[s]
            alias[s] = l
            scaled[l] = (scaled[l] + scaled[s]) - 1.0
            if scaled[l] < 1.0:
                small.append(l)
            else:
                large.append(l)
        for i in large:
            prob[i] = 1.0
        for i in small:
            prob[i] = 1.0
        self.prob = prob
        self.alias = alias

    def sample(self, rng: np.random.Generator, size=None):
        n = self.prob.size
        if size is None:
            i = int(rng.integers(0, n))
            return i if rng.random() < self.prob[i] else int(self.alias[i])
        idx = rng.integers(0, n, size=size)
        accept = rng.random(size=size) < self.prob[idx]
        return np.where(accept, idx, self.alias[idx])
