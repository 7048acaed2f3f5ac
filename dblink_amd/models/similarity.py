"""Attribute similarity functions.

Semantics match the reference (``SimilarityFn.scala:25-107``):

- ``ConstantSimilarityFn``: similarity 0 everywhere.
- ``LevenshteinSimilarityFn(threshold, maxSimilarity)``: truncated, scaled
  normalized Levenshtein similarity (Yujian-Bo normalized metric)::

      unit(a,b)  = 1 - 2*d / (|a| + |b| + d)          (1.0 when both empty)
      sim(a,b)   = max(0, t * (S_max * unit(a,b) - thr)),  t = S_max/(S_max-thr)

These host implementations are the *oracle*; the batched V x V domain sweep
used to build the attribute index runs in the native extension
(CPU C++ / HIP for gfx950) — see ``dblink_amd.ops``.
"""

from __future__ import annotations


class SimilarityFn:
    is_constant = False

    def similarity(self, a: str, b: str) -> float:
        raise NotImplementedError

    def mk_string(self) -> str:
        raise NotImplementedError


class ConstantSimilarityFn(SimilarityFn):
    is_constant = True
    threshold = 0.0
    max_similarity = 0.0

    def similarity(self, a: str, b: str) -> float:
        return 0.0

    def mk_string(self) -> str:
        return "ConstantSimilarityFn"

    def __eq__(self, other):
        return isinstance(other, ConstantSimilarityFn)

    def __hash__(self):
        return hash("ConstantSimilarityFn")


def levenshtein(a: str, b: str) -> int:
    """Plain dynamic-programming edit distance (insert/delete/substitute = 1)."""
    la, lb = len(a), len(b)
    if la == 0:
        return lb
    if lb == 0:
        return la
    prev = list(range(lb + 1))
    cur = [0] * (lb + 1)
    for i in range(1, la + 1):
        cur[0] = i
        ca = a[i - 1]
        for j in range(1, lb + 1):
            cost = 0 if ca == b[j - 1] else 1
            cur[j] = min(prev[j] + 1, cur[j - 1] + 1, prev[j - 1] + cost)
        prev, cur = cur, prev
    return prev[lb]


class LevenshteinSimilarityFn(SimilarityFn):
    is_constant = False

    def __init__(self, threshold: float = 7.0, max_similarity: float = 10.0):
        if max_similarity <= 0.0:
            raise ValueError("`maxSimilarity` must be positive")
        if not (0.0 <= threshold < max_similarity):
            raise ValueError("`threshold` must be in [0, maxSimilarity)")
        self.threshold = float(threshold)
        self.max_similarity = float(max_similarity)
        self.trans_factor = max_similarity / (max_similarity - threshold)

    def unit_similarity(self, a: str, b: str) -> float:
        total = len(a) + len(b)
        if total == 0:
            return 1.0
        d = levenshtein(a, b)
        return 1.0 - 2.0 * d / (total + d)

    def similarity(self, a: str, b: str) -> float:
        s = self.trans_factor * (self.max_similarity * self.unit_similarity(a, b) - self.threshold)
        return s if s > 0.0 else 0.0

    def mk_string(self) -> str:
        return f"LevenshteinSimilarityFn(threshold={self.threshold}, maxSimilarity={self.max_similarity})"

    def __eq__(self, other):
        return (
            isinstance(other, LevenshteinSimilarityFn)
            and other.threshold == self.threshold
            and other.max_similarity == self.max_similarity
        )

    def __hash__(self):
        return hash(("LevenshteinSimilarityFn", self.threshold, self.max_similarity))


def similarity_fn_from_config(cfg) -> SimilarityFn:
    """Build a similarity function from a config object (``Project.scala:203-217``)."""
    name = cfg.get_string("name")
    if name == "ConstantSimilarityFn":
        return ConstantSimilarityFn()
    if name == "LevenshteinSimilarityFn":
        return LevenshteinSimilarityFn(
            threshold=cfg.get_double("parameters.threshold"),
            max_similarity=cfg.get_double("parameters.maxSimilarity"),
        )
    raise ValueError(f"unsupported similarity function: {name!r}")
