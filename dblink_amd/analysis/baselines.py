"""Baseline clusterings (parity: ``analysis/baselines.scala:25-55``)."""

from __future__ import annotations

from collections import defaultdict
from itertools import combinations


def exact_match_clusters(records):
    """records: iterable of (record_id, [values]); cluster by exact value tuple."""
    agg = defaultdict(set)
    for rid, values in records:
        agg[tuple(values)].add(rid)
    return list(agg.values())


def near_clusters(records, num_disagree):
    """Overlapping clusters allowing up to ``num_disagree`` attribute
    disagreements (analysis/baselines.scala:40-54)."""
    if num_disagree < 0:
        raise ValueError("`numDisagree` must be non-negative")
    records = list(records)
    if not records:
        return []
    num_attr = len(records[0][1])
    agg = defaultdict(set)
    for rid, values in records:
        for del_ids in combinations(range(num_attr), num_disagree):
            partial = tuple(v for i, v in enumerate(values) if i not in del_ids)
            agg[(del_ids, partial)].add(rid)
    return list(agg.values())
