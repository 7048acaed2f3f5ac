"""Entity-space partitioning: KD-tree over attribute value ids.

Parity with the reference:
- ``KDTreePartitioner.scala:28-106``: numLevels splits cycling through the
  chosen attributes; each level splits every current leaf on the weighted
  counts of its entities' values for that attribute.
- ``DomainSplitter.scala:30-111``: domains with <= 30 distinct observed
  values use a 2-bin LPT set split; larger domains use a sorted
  weighted-median range split. splitQuality = 1 - |w - half| / half.
- ``MutableBST.scala:33-112``: array-backed tree; leaf numbers are assigned
  in split order (left child keeps the parent's number, right child gets the
  next fresh number).

The fitted tree is exported as flat arrays (``as_flat``) for the GPU descent
kernel: per node {attr, kind, split_value | right-set range} where right-set
members live in one concatenated sorted int32 array.
"""

from __future__ import annotations

import numpy as np


class _Node:
    __slots__ = ("attr_id", "splitter", "value")

    def __init__(self, attr_id, splitter, value):
        self.attr_id = attr_id
        self.splitter = splitter
        self.value = value


class RangeSplitter:
    """Sorted weighted-median split: go right iff value > split_value."""

    def __init__(self, domain):  # domain: list[(value, weight)]
        half = sum(w for _, w in domain) / 2.0
        ordered = sorted(domain)
        cum = 0.0
        i = 0
        while cum <= half and i < len(ordered) - 1:
            cum += ordered[i][1]
            i += 1
        self.split_value = ordered[i][0]
        self.split_weight = cum
        self.split_quality = 1.0 - abs(cum - half) / half if half > 0 else 0.0

    def __call__(self, x):
        return x > self.split_value


class LPTSplitter:
    """2-bin longest-processing-time set split: go right iff value in right set."""

    def __init__(self, domain):
        half = sum(w for _, w in domain) / 2.0
        ordered = sorted(domain, key=lambda kv: -kv[1])
        left_w = 0.0
        right_w = 0.0
        right = set()
        for value, weight in ordered:
            if left_w >= right_w:
                right.add(value)
                right_w += weight
            else:
                left_w += weight
        self.right_set = right
        self.split_quality = 1.0 - abs(left_w - half) / half if half > 0 else 0.0

    def __call__(self, x):
        return x in self.right_set


def make_splitter(domain):
    return LPTSplitter(domain) if len(domain) <= 30 else RangeSplitter(domain)


class MutableBST:
    def __init__(self):
        self.nodes = [_Node(-1, None, 0)]
        self.num_levels = 0
        self.num_leaves = 1

    def leaf_node_id(self, values):
        nid = 0
        while nid < len(self.nodes):
            node = self.nodes[nid]
            if node is not None and node.splitter is not None:
                nid = 2 * nid + 2 if node.splitter(values[node.attr_id]) else 2 * nid + 1
            else:
                return nid
        return nid

    def leaf_number(self, values):
        return self.nodes[self.leaf_node_id(values)].value

    def split_node(self, node_id, attr_id, splitter):
        node = self.nodes[node_id] if node_id < len(self.nodes) else None
        if node is None:
            raise ValueError("node does not exist")
        if node.splitter is not None:
            raise ValueError("node is already split")
        node.attr_id = attr_id
        node.splitter = splitter
        left, right = 2 * node_id + 1, 2 * node_id + 2
        if left >= len(self.nodes):
            self.num_levels += 1
            self.nodes.extend([None] * (2 ** self.num_levels))
        self.nodes[left] = _Node(-1, None, node.value)
        self.nodes[right] = _Node(-1, None, self.num_leaves)
        self.num_leaves += 1


class KDTreePartitioner:
    """fit() over entity attribute-value matrices; get_partition_id per entity."""

    def __init__(self, num_levels: int, attribute_ids):
        if num_levels < 0:
            raise ValueError("numLevels must be non-negative")
        self.num_levels = num_levels
        self.attribute_ids = list(attribute_ids)
        if num_levels > 0 and not self.attribute_ids:
            raise ValueError("attributeIds must be non-empty if numLevels > 0")
        self.tree = MutableBST()

    @property
    def num_partitions(self):
        return self.tree.num_leaves

    def fit(self, values: np.ndarray, log=None):
        """values: int32 [N, A] matrix of entity attribute value ids."""
        values = np.asarray(values)
        N = values.shape[0]
        it = 0
        for level in range(self.num_levels):
            attr_id = self.attribute_ids[it % len(self.attribute_ids)]
            it += 1
            # node id per row under the current tree (descent cache rebuilt
            # each level while the tree grows)
            self._flat_cache = None
            node_ids = self._leaf_node_ids(values)
            col = values[:, attr_id]
            # group (node, value) weights
            order = np.lexsort((col, node_ids))
            sn, sv = node_ids[order], col[order]
            boundaries = np.flatnonzero(np.r_[True, (sn[1:] != sn[:-1]) | (sv[1:] != sv[:-1])])
            counts = np.diff(np.r_[boundaries, len(sn)])
            for nid in np.unique(sn):
                mask = sn[boundaries] == nid
                domain = [
                    (int(sv[b]), float(c)) for b, c in zip(boundaries[mask], counts[mask])
                ]
                splitter = make_splitter(domain)
                if splitter.split_quality <= 0.9 and log is not None:
                    log.warning(
                        "Poor quality split (%.1f%%) at node %d.", splitter.split_quality * 100, nid
                    )
                self.tree.split_node(int(nid), attr_id, splitter)
        self._flat_cache = None
        return self

    def _leaf_node_ids(self, values):
        # flat-array descent (the per-node masking variant re-scanned the
        # whole active set per node and per level); range splits — the
        # common case for string attributes — are fully vectorized, set
        # splits use a per-node sorted-slice binary search
        flat = getattr(self, "_flat_cache", None)
        if flat is None:
            flat = self.as_flat()
            self._flat_cache = flat
        kind, attr, a, b, rset = (flat["kind"], flat["attr"], flat["a"],
                                  flat["b"], flat["rset"])
        N = values.shape[0]
        node = np.zeros(N, dtype=np.int64)
        for _ in range(self.num_levels):
            k = kind[node]
            act = k != 0
            if not act.any():
                break
            right = np.zeros(N, dtype=bool)
            rsel = k == 1
            if rsel.any():
                col = values[rsel, attr[node[rsel]]]
                right[rsel] = col > a[node[rsel]]
            ssel = k == 2
            if ssel.any():
                idx = np.flatnonzero(ssel)
                nids = node[idx]
                col = values[idx, attr[nids]]
                for nid in np.unique(nids):
                    m = nids == nid
                    lo, ln = int(a[nid]), int(b[nid])
                    if ln == 0:
                        continue
                    sl = rset[lo:lo + ln]
                    cm = col[m]
                    p = np.searchsorted(sl, cm)
                    right[idx[m]] = (p < ln) & (sl[np.minimum(p, ln - 1)] == cm)
            node = np.where(act, np.where(right, 2 * node + 2, 2 * node + 1),
                            node)
        return node

    def get_partition_id(self, values) -> int:
        return self.tree.leaf_number(values)

    def get_partition_ids(self, values: np.ndarray) -> np.ndarray:
        """Vectorized leaf numbers for an [N, A] matrix."""
        node_ids = self._leaf_node_ids(np.asarray(values))
        leaf_vals = np.array(
            [n.value if n is not None else 0 for n in self.tree.nodes], dtype=np.int64
        )
        return leaf_vals[node_ids].astype(np.int32)

    def mk_string(self):
        if self.num_levels == 0:
            return "KDTreePartitioner(numLevels=0)"
        ids = ",".join(str(i) for i in self.attribute_ids)
        return f"KDTreePartitioner(numLevels={self.num_levels}, attributeIds=[{ids}])"

    # ---- flat export for the GPU descent kernel ------------------------------

    def as_flat(self):
        """Flat arrays: for node i —
        kind[i]: 0 = leaf, 1 = range split, 2 = set split
        attr[i]: split attribute (or -1)
        a[i]:    split_value (range) or right-set start offset (set) or leaf number
        b[i]:    right-set length (set) else 0
        rset:    concatenated sorted right-set value ids
        """
        n = len(self.tree.nodes)
        kind = np.zeros(n, dtype=np.int32)
        attr = np.full(n, -1, dtype=np.int32)
        a = np.zeros(n, dtype=np.int32)
        b = np.zeros(n, dtype=np.int32)
        rset = []
        for i, node in enumerate(self.tree.nodes):
            if node is None:
                kind[i] = 0
                a[i] = 0
                continue
            if node.splitter is None:
                kind[i] = 0
                a[i] = node.value if node.value is not None else 0
            elif isinstance(node.splitter, RangeSplitter):
                kind[i] = 1
                attr[i] = node.attr_id
                a[i] = int(node.splitter.split_value)
            else:
                kind[i] = 2
                attr[i] = node.attr_id
                members = np.array(sorted(node.splitter.right_set), dtype=np.int32)
                a[i] = int(sum(len(x) for x in rset))
                b[i] = len(members)
                rset.append(members)
        rset_arr = np.concatenate(rset) if rset else np.empty(0, dtype=np.int32)
        return {"kind": kind, "attr": attr, "a": a, "b": b, "rset": rset_arr}


def partitioner_from_config(cfg, attribute_names):
    """``Project.scala:219-229``. KDTreePartitioner is the reference surface;
    SimplePartitioner is additionally accepted (API-only in the reference)."""
    name = cfg.get_string("name")
    if name == "KDTreePartitioner":
        num_levels = cfg.get_int("parameters.numLevels")
        names = cfg.get_string_list("parameters.matchingAttributes")
        attr_ids = [list(attribute_names).index(n) for n in names]
        return KDTreePartitioner(num_levels, attr_ids)
    if name == "SimplePartitioner":
        attr = cfg.get_string("parameters.attribute")
        return SimplePartitioner(
            list(attribute_names).index(attr), cfg.get_int("parameters.numPartitions")
        )
    raise ValueError("unsupported partitioner: " + name)


class LPTScheduler:
    """Longest-processing-time greedy assignment of weighted jobs to
    partitions (parity: ``partitioning/LPTScheduler.scala:38-85``)."""

    def __init__(self, jobs, num_partitions):
        """jobs: list of (job_key, weight)."""
        if num_partitions <= 0:
            raise ValueError("numPartitions must be positive")
        self.num_partitions = num_partitions
        loads = [0.0] * num_partitions
        assignment = {}
        for key, w in sorted(jobs, key=lambda kv: -kv[1]):
            p = min(range(num_partitions), key=lambda i: loads[i])
            loads[p] += w
            assignment[key] = p
        self.assignment = assignment
        self.loads = loads

    def partition_of(self, key):
        return self.assignment[key]


class SimplePartitioner:
    """Blocks on a single attribute's value, LPT bin-packed into
    ``num_partitions`` (parity: ``partitioning/SimplePartitioner.scala:33-64``;
    API-only in the reference — not reachable from its config parser)."""

    def __init__(self, attribute_id: int, num_partitions: int):
        self.attribute_id = attribute_id
        self._num_partitions = num_partitions
        self.scheduler = None

    @property
    def num_partitions(self):
        return self._num_partitions

    def fit(self, values: np.ndarray, log=None):
        col = np.asarray(values)[:, self.attribute_id]
        uniq, counts = np.unique(col, return_counts=True)
        self.scheduler = LPTScheduler(
            list(zip(uniq.tolist(), counts.astype(float).tolist())), self._num_partitions
        )
        return self

    def get_partition_id(self, values) -> int:
        v = values[self.attribute_id]
        return self.scheduler.assignment.get(int(v), 0)

    def get_partition_ids(self, values: np.ndarray) -> np.ndarray:
        col = np.asarray(values)[:, self.attribute_id]
        out = np.zeros(col.shape[0], dtype=np.int32)
        for i, v in enumerate(col):
            out[i] = self.scheduler.assignment.get(int(v), 0)
        return out

    def mk_string(self):
        return (
            f"SimplePartitioner(attributeId={self.attribute_id}, "
            f"numPartitions={self._num_partitions})"
        )

    def as_flat(self):
        """Flat export: emulated as a single-level 'set' table is not possible
        (arbitrary value -> partition map); the GPU engine detects the dense
        map attribute instead."""
        raise NotImplementedError(
            "SimplePartitioner has no flat KD export; use value_map()"
        )

    def value_map(self, num_values: int) -> np.ndarray:
        out = np.zeros(num_values, dtype=np.int32)
        for v, p in self.scheduler.assignment.items():
            if 0 <= v < num_values:
                out[v] = p
        return out
