import numpy as np
import pytest

from dblink_amd.parallel.partitioning import (
    KDTreePartitioner,
    LPTSplitter,
    RangeSplitter,
    make_splitter,
)


def test_range_splitter_median():
    domain = [(i, 1.0) for i in range(100)]
    s = RangeSplitter(domain)
    assert 0.9 <= s.split_quality <= 1.0
    left = sum(1 for v, _ in domain if not s(v))
    right = sum(1 for v, _ in domain if s(v))
    # the reference's splitter takes the first value past the weighted median,
    # so the split can be off-centre by a few values (52/48 on 100 uniform)
    assert abs(left - right) <= 4


def test_lpt_splitter_balance():
    domain = [(i, w) for i, w in enumerate([5.0, 4.0, 3.0, 2.0, 1.0, 1.0])]
    s = LPTSplitter(domain)
    left = sum(w for v, w in domain if not s(v))
    right = sum(w for v, w in domain if s(v))
    assert abs(left - right) <= 2.0
    assert s.split_quality > 0.7


def test_make_splitter_dispatch():
    small = [(i, 1.0) for i in range(10)]
    large = [(i, 1.0) for i in range(100)]
    assert isinstance(make_splitter(small), LPTSplitter)
    assert isinstance(make_splitter(large), RangeSplitter)


def test_kdtree_zero_levels():
    p = KDTreePartitioner(0, [])
    vals = np.random.default_rng(0).integers(0, 50, size=(200, 3)).astype(np.int32)
    p.fit(vals)
    assert p.num_partitions == 1
    assert np.all(p.get_partition_ids(vals) == 0)


@pytest.mark.parametrize("levels,expected", [(1, 2), (2, 4), (3, 8)])
def test_kdtree_levels(levels, expected):
    rng = np.random.default_rng(0)
    vals = rng.integers(0, 1000, size=(5000, 2)).astype(np.int32)
    p = KDTreePartitioner(levels, [0, 1])
    p.fit(vals)
    assert p.num_partitions == expected
    pids = p.get_partition_ids(vals)
    assert set(np.unique(pids)) == set(range(expected))
    # balanced within ~25%
    counts = np.bincount(pids, minlength=expected)
    assert counts.min() > 0.6 * counts.mean()


def test_kdtree_flat_descent_matches():
    """Flat-array export must agree with the object-tree descent."""
    rng = np.random.default_rng(1)
    vals = rng.integers(0, 20, size=(2000, 3)).astype(np.int32)  # small domain -> LPT splits
    p = KDTreePartitioner(2, [0, 2])
    p.fit(vals)
    flat = p.as_flat()
    ref = p.get_partition_ids(vals)

    # simulate descent via flat arrays (mirrors the HIP kernel logic)
    def descend(row):
        nid = 0
        while flat["kind"][nid] != 0:
            a = flat["attr"][nid]
            v = row[a]
            if flat["kind"][nid] == 1:
                right = v > flat["a"][nid]
            else:
                lo, n = flat["a"][nid], flat["b"][nid]
                members = flat["rset"][lo : lo + n]
                right = np.searchsorted(members, v) < n and members[np.searchsorted(members, v)] == v
            nid = 2 * nid + 2 if right else 2 * nid + 1
        return flat["a"][nid]

    got = np.array([descend(vals[i]) for i in range(0, 2000, 37)])
    np.testing.assert_array_equal(got, ref[::37])


def test_kdtree_deterministic():
    vals = np.random.default_rng(3).integers(0, 500, size=(3000, 2)).astype(np.int32)
    p1 = KDTreePartitioner(2, [0, 1]).fit(vals)
    p2 = KDTreePartitioner(2, [0, 1]).fit(vals)
    np.testing.assert_array_equal(p1.get_partition_ids(vals), p2.get_partition_ids(vals))


def test_lpt_scheduler():
    from dblink_amd.parallel.partitioning import LPTScheduler

    jobs = [("a", 10.0), ("b", 8.0), ("c", 6.0), ("d", 5.0), ("e", 4.0), ("f", 3.0)]
    s = LPTScheduler(jobs, 2)
    assert abs(s.loads[0] - s.loads[1]) <= 2.0
    assert set(s.assignment) == {"a", "b", "c", "d", "e", "f"}


def test_simple_partitioner():
    from dblink_amd.parallel.partitioning import SimplePartitioner

    rng = np.random.default_rng(0)
    vals = rng.integers(0, 12, size=(1000, 2)).astype(np.int32)
    p = SimplePartitioner(1, 4).fit(vals)
    pids = p.get_partition_ids(vals)
    assert set(np.unique(pids)) <= set(range(4))
    # same value always lands on the same partition (blocking invariant)
    for v in range(12):
        rows = vals[:, 1] == v
        if rows.any():
            assert len(set(pids[rows])) == 1
    counts = np.bincount(pids, minlength=4)
    assert counts.max() < 2.2 * counts.mean()


def test_kd_partitioner_randomized():
    """Property test: fitted KD partition ids stay in range (seen and unseen
    values), and the flat export descends to the same leaf as the tree."""
    from hypothesis import given, settings
    from hypothesis import strategies as st

    @settings(max_examples=40, deadline=None)
    @given(
        n=st.integers(min_value=2, max_value=300),
        a=st.integers(min_value=1, max_value=4),
        levels=st.integers(min_value=0, max_value=4),
        vmax=st.integers(min_value=2, max_value=40),
        seed=st.integers(min_value=0, max_value=2**31),
    )
    def check(n, a, levels, vmax, seed):
        rng = np.random.default_rng(seed)
        vals = rng.integers(0, vmax, size=(n, a)).astype(np.int32)
        p = KDTreePartitioner(levels, list(range(a)) * max(1, levels))
        p.fit(vals)
        pids = p.get_partition_ids(vals)
        assert pids.min() >= 0 and pids.max() < p.num_partitions <= 2 ** levels
        flat = p.as_flat()

        def descend(row):
            nid = 0
            while flat["kind"][nid] != 0:
                attr = flat["attr"][nid]
                if flat["kind"][nid] == 1:
                    go_right = row[attr] > flat["a"][nid]
                else:
                    lo, ln = flat["a"][nid], flat["b"][nid]
                    go_right = row[attr] in set(flat["rset"][lo:lo + ln].tolist())
                nid = 2 * nid + (2 if go_right else 1)
            return flat["a"][nid]

        for i in rng.integers(0, n, min(15, n)):
            assert descend(vals[i]) == pids[i]
        unseen = rng.integers(0, vmax + 5, size=(8, a)).astype(np.int32)
        up = p.get_partition_ids(unseen)
        assert up.min() >= 0 and up.max() < p.num_partitions

    check()
