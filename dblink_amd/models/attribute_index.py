"""Attribute domain index: value ids, empirical distribution, sparse
similarity index, normalizers and cached power distributions.

Mirrors the reference's ``AttributeIndex.scala:39-246`` capability surface:

- value ids are assigned by sorting the distinct strings lexicographically
  (``AttributeIndex.scala:113-116``)
- the similarity index stores, per value id v, the sparse row
  ``{w : exp(sim(v,w))}`` restricted to exp(sim) > 1 (``:219-231``);
  ``exp_sim_of`` returns 1.0 for absent pairs (``:183-186``)
- ``sim_normalization_of(v)`` returns ``1 / sum_w phi(w)*exp(sim(w,v))``
  (``:234-245``)
- ``sim_norm_dist(k)`` is the distribution ``p(v) ∝ phi(v)*norm(v)^k``,
  pre-cached for k = 1..max_cluster_size (``:188-216``)

Unlike the reference (a Spark ``cartesian`` over V x V pairs), the pair sweep
here is a length-bucketed, threshold-pruned batched edit-distance pass that
runs in the native extension (HIP on gfx950, C++/OpenMP on host) — see
``dblink_amd.ops.sim_pairs``.
"""

from __future__ import annotations

import math
from bisect import bisect_left

import numpy as np

from .alias import AliasTable
from .similarity import SimilarityFn


class SimIndexCSR:
    """Sparse V x V exp-similarity matrix in CSR form (rows sorted by col id)."""

    __slots__ = ("row_ptr", "col", "expsim")

    def __init__(self, row_ptr: np.ndarray, col: np.ndarray, expsim: np.ndarray):
        self.row_ptr = np.asarray(row_ptr, dtype=np.int64)
        self.col = np.asarray(col, dtype=np.int32)
        self.expsim = np.asarray(expsim, dtype=np.float64)

    def row(self, v: int):
        lo, hi = self.row_ptr[v], self.row_ptr[v + 1]
        return self.col[lo:hi], self.expsim[lo:hi]

    def lookup(self, v: int, w: int) -> float:
        lo, hi = int(self.row_ptr[v]), int(self.row_ptr[v + 1])
        i = lo + bisect_left(self.col[lo:hi].tolist(), w)
        if i < hi and self.col[i] == w:
            return float(self.expsim[i])
        return 1.0

    def lookup_row_many(self, v: int, ws) -> np.ndarray:
        """Vectorized ``lookup(v, w)`` for an array of ``w`` (same values)."""
        lo, hi = int(self.row_ptr[v]), int(self.row_ptr[v + 1])
        cols = self.col[lo:hi]
        idx = np.searchsorted(cols, ws)
        idx_c = np.minimum(idx, hi - lo - 1) if hi > lo else idx * 0
        hit = (hi > lo) & (cols[idx_c] == ws)
        out = np.ones(len(ws), dtype=np.float64)
        out[hit] = self.expsim[lo + idx_c[hit]]
        return out

    @property
    def nnz(self):
        return int(self.col.size)


def _python_sim_pairs(values, similarity_fn):
    """Reference pair sweep in pure Python (test oracle / tiny domains)."""
    rows = [[] for _ in values]
    for i, a in enumerate(values):
        for j, b in enumerate(values):
            s = similarity_fn.similarity(a, b)
            e = math.exp(s)
            if e > 1.0:
                rows[i].append((j, e))
    row_ptr = np.zeros(len(values) + 1, dtype=np.int64)
    cols, sims = [], []
    for i, row in enumerate(rows):
        row.sort()
        row_ptr[i + 1] = row_ptr[i] + len(row)
        for j, e in row:
            cols.append(j)
            sims.append(e)
    return SimIndexCSR(row_ptr, np.asarray(cols, dtype=np.int32), np.asarray(sims, dtype=np.float64))


class AttributeIndex:
    """Index over one attribute's domain."""

    def __init__(
        self,
        values_weights: dict,
        similarity_fn: SimilarityFn,
        precache_powers: int = 0,
        pair_sweep=None,
    ):
        if not values_weights:
            raise ValueError("index cannot be empty")
        items = sorted(values_weights.items())
        self.values = [k for k, _ in items]
        weights = np.array([w for _, w in items], dtype=np.float64)
        total = weights.sum()
        self.probs = weights / total
        self._string_to_id = {v: i for i, v in enumerate(self.values)}
        self.similarity_fn = similarity_fn
        self.is_constant = similarity_fn.is_constant

        self._distribution = AliasTable(self.probs)

        if self.is_constant:
            self.sim_index = None
            self.sim_norms = np.ones(self.num_values, dtype=np.float64)
        else:
            if pair_sweep is None:
                from .. import ops

                pair_sweep = ops.sim_pairs
            self.sim_index = pair_sweep(self.values, similarity_fn)
            self.sim_norms = self._compute_normalizations()

        self._max_cached_power = 0
        self._power_dists = {}
        self._power_totals = {}
        if not self.is_constant:
            for k in range(1, precache_powers + 1):
                self._build_power(k)
            self._max_cached_power = precache_powers

    # ---- construction helpers -------------------------------------------------

    def _compute_normalizations(self):
        # norm_v = 1 / sum_w phi(w) * expsim(w, v); expsim symmetric, rows of v
        # hold exactly the pairs with expsim > 1 (AttributeIndex.scala:234-245).
        sums = np.full(self.num_values, 0.0)
        base = float(self.probs.sum())  # contributions of expsim == 1 pairs
        extra = np.zeros(self.num_values, dtype=np.float64)
        rp, col, es = self.sim_index.row_ptr, self.sim_index.col, self.sim_index.expsim
        for v in range(self.num_values):
            lo, hi = rp[v], rp[v + 1]
            if hi > lo:
                c = col[lo:hi]
                extra[v] = np.sum(self.probs[c] * (es[lo:hi] - 1.0))
        sums = base + extra
        return 1.0 / sums

    def _build_power(self, k: int):
        w = self.probs * np.power(self.sim_norms, k)
        self._power_dists[k] = AliasTable(w)
        self._power_totals[k] = float(w.sum())

    # ---- query API (AttributeIndex.scala trait) -------------------------------

    @property
    def num_values(self) -> int:
        return len(self.values)

    @property
    def log_probs(self):
        """log(phi), cached: the per-sweep summary reduction gathers these
        for every entity and distorted record."""
        lp = getattr(self, "_log_probs", None)
        if lp is None:
            lp = np.log(self.probs)
            self._log_probs = lp
        return lp

    @property
    def log_sim_norms(self):
        lsn = getattr(self, "_log_sim_norms", None)
        if lsn is None:
            lsn = np.log(self.sim_norms)
            self._log_sim_norms = lsn
        return lsn

    @property
    def distribution(self) -> AliasTable:
        return self._distribution

    def probability_of(self, value_id: int) -> float:
        if not (0 <= value_id < self.num_values):
            raise IndexError("valueId is not in the index")
        return float(self.probs[value_id])

    def draw(self, rng) -> int:
        return int(self._distribution.sample(rng))

    def value_id_of(self, value: str) -> int:
        return self._string_to_id.get(value, -1)

    def sim_normalization_of(self, value_id: int) -> float:
        if not (0 <= value_id < self.num_values):
            raise IndexError("valueId is not in the index")
        return float(self.sim_norms[value_id])

    def sim_values_of(self, value_id: int):
        """Sparse row {similar value id -> expsim} (empty map for constant)."""
        if not (0 <= value_id < self.num_values):
            raise IndexError("valueId is not in the index")
        if self.is_constant:
            return {}
        cols, sims = self.sim_index.row(value_id)
        return dict(zip(cols.tolist(), sims.tolist()))

    def exp_sim_of(self, v1: int, v2: int) -> float:
        if not (0 <= v1 < self.num_values):
            raise IndexError("valueId1 is not in the index")
        if not (0 <= v2 < self.num_values):
            raise IndexError("valueId2 is not in the index")
        if self.is_constant:
            return 1.0
        return self.sim_index.lookup(v1, v2)

    def sim_row_len(self, v: int) -> int:
        if self.is_constant:
            return 0
        return int(self.sim_index.row_ptr[v + 1] - self.sim_index.row_ptr[v])

    @property
    def self_mass(self) -> np.ndarray:
        """phi(v) [* norm(v) * expsim(v, v)] — the agreement mass used by the
        distortion conditional (GibbsUpdates.scala:332-346); static, cached."""
        got = getattr(self, "_self_mass", None)
        if got is None:
            if self.is_constant:
                got = self.probs.copy()
            else:
                es = np.array([self.exp_sim_of(v, v) for v in range(self.num_values)])
                got = self.probs * self.sim_norms * es
            self._self_mass = got
        return got

    def exp_sim_many(self, v1: int, v2s) -> np.ndarray:
        """Vectorized ``exp_sim_of(v1, .)`` over an int array (same values)."""
        if self.is_constant:
            return np.ones(len(v2s))
        return self.sim_index.lookup_row_many(v1, np.asarray(v2s))

    def sim_norm_dist(self, power: int) -> AliasTable:
        """Distribution p(v) ∝ phi(v) * norm(v)^power."""
        if power <= 0:
            raise ValueError("power must be a positive integer")
        if self.is_constant:
            return self._distribution
        if power not in self._power_dists:
            self._build_power(power)
        return self._power_dists[power]

    def sim_norm_total(self, power: int) -> float:
        """Total unnormalized weight of ``sim_norm_dist(power)``."""
        if self.is_constant:
            return 1.0
        if power not in self._power_totals:
            self._build_power(power)
        return self._power_totals[power]

    def sim_norm_prob(self, value_id: int, power: int) -> float:
        """probabilityOf under ``sim_norm_dist(power)`` (normalized)."""
        if self.is_constant:
            return self.probability_of(value_id)
        return (
            self.probs[value_id] * self.sim_norms[value_id] ** power / self.sim_norm_total(power)
        )
