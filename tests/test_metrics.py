import pytest

from dblink_amd.analysis.baselines import exact_match_clusters, near_clusters
from dblink_amd.analysis.metrics import (
    ClusteringMetrics,
    PairwiseMetrics,
    adjusted_rand_index,
    membership_to_clusters,
    to_membership,
    to_pairwise_links,
)


def test_pairwise_links():
    clusters = [{"a", "b", "c"}, {"d"}, {"e", "f"}]
    links = to_pairwise_links(clusters)
    assert links == {("a", "b"), ("a", "c"), ("b", "c"), ("e", "f")}


def test_pairwise_metrics_perfect():
    clusters = [{"a", "b"}, {"c", "d", "e"}]
    m = PairwiseMetrics.compute(clusters, clusters)
    assert m.precision == 1.0 and m.recall == 1.0 and m.f1score == 1.0


def test_pairwise_metrics_partial():
    pred = [{"a", "b"}, {"c"}, {"d", "e"}]
    true = [{"a", "b", "c"}, {"d"}, {"e"}]
    # pred links: ab, de ; true links: ab, ac, bc
    m = PairwiseMetrics.compute(pred, true)
    assert m.precision == pytest.approx(1 / 2)
    assert m.recall == pytest.approx(1 / 3)
    assert m.f1score == pytest.approx(2 * (1 / 2) * (1 / 3) / (1 / 2 + 1 / 3))


def test_ari_identical():
    clusters = [{"a", "b"}, {"c"}, {"d", "e", "f"}]
    assert adjusted_rand_index(clusters, clusters) == pytest.approx(1.0)


def test_ari_known_value():
    # Hand-computed: contingency nij = [[2,1],[1,2]], n=6 ->
    # sum comb2(nij)=2, pred/true comb sums = 6, expected = 36/15 = 2.4,
    # ARI = (2-2.4)/(6-2.4) = -1/9.
    pred = membership_to_clusters({i: l for i, l in enumerate([0, 0, 1, 1, 0, 1])})
    true = membership_to_clusters({i: l for i, l in enumerate([0, 0, 0, 1, 1, 1])})
    ari = adjusted_rand_index(pred, true)
    assert ari == pytest.approx(-1 / 9, abs=1e-9)

    # cross-check against sklearn if available
    try:
        from sklearn.metrics import adjusted_rand_score
    except ImportError:
        return
    assert ari == pytest.approx(adjusted_rand_score([0, 0, 1, 1, 0, 1], [0, 0, 0, 1, 1, 1]))


def test_ari_mismatched_elements():
    with pytest.raises(ValueError):
        adjusted_rand_index([{"a"}], [{"b"}])


def test_membership_roundtrip():
    clusters = [{"a", "b"}, {"c"}]
    m = to_membership(clusters)
    back = membership_to_clusters(m)
    assert sorted(map(sorted, back)) == sorted(map(sorted, clusters))


def test_exact_match_clusters():
    recs = [("r1", ["x", "y"]), ("r2", ["x", "y"]), ("r3", ["x", "z"])]
    clusters = exact_match_clusters(recs)
    assert sorted(map(sorted, clusters)) == [["r1", "r2"], ["r3"]]


def test_near_clusters():
    recs = [("r1", ["x", "y"]), ("r2", ["x", "z"]), ("r3", ["w", "w"])]
    clusters = near_clusters(recs, 1)
    merged = [c for c in clusters if len(c) > 1]
    assert {"r1", "r2"} in merged


def test_clustering_metrics_report():
    pred = [frozenset({1, 2}), frozenset({3})]
    true = [frozenset({1, 2, 3})]
    m = ClusteringMetrics.compute(pred, true)
    assert m.adj_rand_index == pytest.approx(adjusted_rand_index(pred, true))
    s = m.mk_string()
    assert "Adj. Rand index" in s and str(m.adj_rand_index) in s


def test_ari_and_f1_invariances_randomized():
    """Property test: ARI is invariant to cluster relabeling/permutation and
    equals 1 iff the clusterings match; pairwise F1 is symmetric in its
    confusion counts."""
    from hypothesis import given, settings
    from hypothesis import strategies as st

    from dblink_amd.analysis.metrics import (
        PairwiseMetrics,
        to_pairwise_links,
    )

    @settings(max_examples=40, deadline=None)
    @given(
        membership=st.lists(st.integers(min_value=0, max_value=4),
                            min_size=3, max_size=24),
        seed=st.integers(min_value=0, max_value=2**31),
    )
    def check(membership, seed):
        import numpy as np

        rng = np.random.default_rng(seed)
        items = list(range(len(membership)))
        clusters = membership_to_clusters(dict(zip(items, membership)))
        # relabel + permute items inside clusters
        perm = rng.permutation(len(membership))
        relabeled = membership_to_clusters(
            {i: 10 + membership[i] for i in items}
        )
        n_multi = sum(1 for c in clusters if len(c) > 1)
        if 1 < len(clusters) <= len(membership) - 1 or (n_multi and len(clusters) > 1):
            # ARI is 0/0-undefined for the trivial clusterings (all one
            # cluster / all singletons) — same as the reference's formula
            assert adjusted_rand_index(clusters, relabeled) == pytest.approx(1.0)
        # a second random clustering: ARI symmetric
        other_m = rng.integers(0, 3, len(membership))
        other = membership_to_clusters(dict(zip(items, other_m.tolist())))
        try:
            a1 = adjusted_rand_index(clusters, other)
            a2 = adjusted_rand_index(other, clusters)
            assert a1 == pytest.approx(a2)
            assert a1 <= 1.0 + 1e-12
        except ZeroDivisionError:
            pass  # degenerate all-singleton-vs-all-singleton corner
        if to_pairwise_links(clusters) and to_pairwise_links(other):
            pm1 = PairwiseMetrics.compute(clusters, other)
            pm2 = PairwiseMetrics.compute(other, clusters)
            assert pm1.precision == pytest.approx(pm2.recall)
            assert pm1.recall == pytest.approx(pm2.precision)
        assert to_pairwise_links(clusters) == to_pairwise_links(relabeled)

    check()


def test_distortion_probs_unit():
    """Prior-mean init, call semantics, out-of-range errors, Beta update
    bounds (DistortionProbsTest.scala:24-46 analog)."""
    import numpy as np

    from dblink_amd.models.distortion import DistortionProbs, update_dist_probs
    from dblink_amd.models.records import BetaShapeParameters

    priors = [BetaShapeParameters(1.0, 9.0), BetaShapeParameters(0.5, 49.5)]
    dp = DistortionProbs.from_prior_mean(priors, num_files=3)
    assert dp(0, 0) == pytest.approx(0.1)
    assert dp(1, 2) == pytest.approx(0.01)
    with pytest.raises(IndexError):
        dp(5, 0)
    with pytest.raises(IndexError):
        dp(0, 7)
    rng = np.random.default_rng(0)
    agg = np.array([[3, 0, 1], [0, 0, 0]], dtype=np.int64)
    sizes = np.array([10, 10, 10], dtype=np.int64)
    out = update_dist_probs(agg, priors, sizes, rng)
    assert out.probs.shape == (2, 3)
    assert ((out.probs > 0) & (out.probs < 1)).all()
    # more observed distortions pull theta up
    many = update_dist_probs(np.array([[9, 0, 0], [0, 0, 0]]), priors, sizes,
                             np.random.default_rng(1))
    assert many.probs[0, 0] > out.probs[0, 1]
