"""Attribute-index tests with golden values from the reference's
AttributeIndexTest.scala:38-100 (Australian state names fixture)."""

import numpy as np
import pytest

from dblink_amd.models.attribute_index import AttributeIndex, _python_sim_pairs
from dblink_amd.models.similarity import ConstantSimilarityFn, LevenshteinSimilarityFn

STATE_WEIGHTS = {
    "Australian Capital Territory": 0.410,
    "New South Wales": 7.86,
    "Northern Territory": 0.246,
    "Queensland": 4.92,
    "South Australia": 1.72,
    "Tasmania": 0.520,
    "Victoria": 6.32,
    "Western Australia": 2.58,
}

# Golden sim normalizations from AttributeIndexTest.scala:48-56
# (LevenshteinSimilarityFn(5.0, 10.0))
GOLDEN_NORMS = {
    "Australian Capital Territory": 0.0027140755302269004,
    "New South Wales": 1.4193905286944585e-4,
    "Northern Territory": 0.00451528932619675,
    "Queensland": 2.2673706056780077e-4,
    "South Australia": 6.465919296781136e-4,
    "Tasmania": 0.00214117348291189,
    "Victoria": 1.7651936247903708e-4,
    "Western Australia": 4.317863538883541e-4,
}


@pytest.fixture(scope="module")
def const_index():
    return AttributeIndex(STATE_WEIGHTS, ConstantSimilarityFn(), pair_sweep=_python_sim_pairs)


@pytest.fixture(scope="module")
def lev_index():
    return AttributeIndex(
        STATE_WEIGHTS, LevenshteinSimilarityFn(5.0, 10.0), pair_sweep=_python_sim_pairs
    )


def test_value_ids_sorted(const_index):
    # ids assigned by lexicographic sort of distinct values
    assert const_index.value_id_of("Australian Capital Territory") == 0
    assert const_index.value_id_of("Western Australia") == 7
    assert const_index.value_id_of("not a state") == -1


def test_probabilities(const_index):
    total = sum(STATE_WEIGHTS.values())
    for name, w in STATE_WEIGHTS.items():
        vid = const_index.value_id_of(name)
        assert const_index.probability_of(vid) == pytest.approx(w / total)
    assert sum(const_index.probs) == pytest.approx(1.0)


def test_probability_out_of_range(const_index):
    with pytest.raises(IndexError):
        const_index.probability_of(-1)
    with pytest.raises(IndexError):
        const_index.probability_of(const_index.num_values)


def test_constant_norms_and_sims(const_index):
    for v in range(const_index.num_values):
        assert const_index.sim_normalization_of(v) == 1.0
        assert const_index.sim_values_of(v) == {}
        for w in range(const_index.num_values):
            assert const_index.exp_sim_of(v, w) == 1.0


def test_golden_sim_normalizations(lev_index):
    for name, golden in GOLDEN_NORMS.items():
        vid = lev_index.value_id_of(name)
        assert lev_index.sim_normalization_of(vid) == pytest.approx(golden, abs=1e-4)


def test_golden_sim_values_sa(lev_index):
    # AttributeIndexTest.scala:58: simValuesOf("South Australia") has exactly
    # {7 -> 39.813678..., 4 -> 22026.4657...} (ids 7=WA, 4=SA itself)
    vid = lev_index.value_id_of("South Australia")
    sims = lev_index.sim_values_of(vid)
    assert set(sims.keys()) == {4, 7}
    assert sims[7] == pytest.approx(39.813678188084864, rel=1e-6)
    assert sims[4] == pytest.approx(22026.465794806718, rel=1e-6)


def test_golden_exp_sims(lev_index):
    sa = lev_index.value_id_of("South Australia")
    wa = lev_index.value_id_of("Western Australia")
    assert lev_index.exp_sim_of(sa, wa) == pytest.approx(39.813678188084864, rel=1e-6)
    vic = lev_index.value_id_of("Victoria")
    tas = lev_index.value_id_of("Tasmania")
    assert lev_index.exp_sim_of(vic, tas) == 1.0


def test_power_distribution(lev_index):
    d1 = lev_index.sim_norm_dist(1)
    # p(v) proportional to phi(v) * norm(v)
    expect = lev_index.probs * lev_index.sim_norms
    expect /= expect.sum()
    np.testing.assert_allclose(d1.probs, expect, rtol=1e-12)
    # sim_norm_prob matches
    for v in range(lev_index.num_values):
        assert lev_index.sim_norm_prob(v, 1) == pytest.approx(expect[v])
    with pytest.raises(ValueError):
        lev_index.sim_norm_dist(0)


def test_constant_power_dist_is_phi(const_index):
    d = const_index.sim_norm_dist(3)
    np.testing.assert_allclose(d.probs, const_index.probs)


def test_native_pair_sweep_matches_python():
    """If the native extension is built, its sim_pairs must agree with the
    pure-Python oracle on this fixture."""
    from dblink_amd import ops

    if not ops.have_native():
        pytest.skip("native extension not built")
    fn = LevenshteinSimilarityFn(5.0, 10.0)
    values = sorted(STATE_WEIGHTS)
    native = ops.sim_pairs(values, fn)
    ref = _python_sim_pairs(values, fn)
    np.testing.assert_array_equal(native.row_ptr, ref.row_ptr)
    np.testing.assert_array_equal(native.col, ref.col)
    np.testing.assert_allclose(native.expsim, ref.expsim, rtol=1e-6)


def test_native_pair_sweep_matches_python_randomized():
    """Property test: native (OpenMP) pair sweep vs the pure-Python oracle on
    random ASCII domains and thresholds."""
    from hypothesis import given, settings
    from hypothesis import strategies as st

    from dblink_amd import ops

    if not ops.have_native():
        pytest.skip("native extension not built")

    alphabet = "ABCDEF \u00dc\u00d6\u0141"  # incl. non-ASCII (char-level distance)

    @settings(max_examples=25, deadline=None)
    @given(
        values=st.lists(st.text(alphabet=alphabet, min_size=0, max_size=14),
                        min_size=2, max_size=24, unique=True),
        threshold=st.floats(min_value=0.5, max_value=9.5),
        max_sim=st.floats(min_value=1.0, max_value=12.0),
    )
    def check(values, threshold, max_sim):
        if threshold >= max_sim:
            threshold = max_sim * 0.7
        fn = LevenshteinSimilarityFn(threshold, max_sim)
        values = sorted(values)
        native = ops.sim_pairs(values, fn)
        ref = _python_sim_pairs(values, fn)
        np.testing.assert_array_equal(native.row_ptr, ref.row_ptr)
        np.testing.assert_array_equal(native.col, ref.col)
        np.testing.assert_allclose(native.expsim, ref.expsim, rtol=1e-6)

    check()


def test_build_inverted_index_unit():
    """(attr, value) -> ascending entity lists, duplicate values grouped
    (parity: EntityInvertedIndexTest.scala:6-37)."""
    from dblink_amd.engine.cpu_engine import _build_inverted_index

    ev = np.array(
        [[0, 5], [1, 5], [0, 6], [2, 5], [0, 5]],
        dtype=np.int32,
    )
    inv = _build_inverted_index(ev)
    np.testing.assert_array_equal(inv[(0, 0)], [0, 2, 4])
    np.testing.assert_array_equal(inv[(0, 1)], [1])
    np.testing.assert_array_equal(inv[(0, 2)], [3])
    np.testing.assert_array_equal(inv[(1, 5)], [0, 1, 3, 4])
    np.testing.assert_array_equal(inv[(1, 6)], [2])
    assert (0, 3) not in inv


def test_native_pair_sweep_wide_charset_fallback():
    """Domains with > 255 distinct characters take the UTF-8 byte pass —
    it must still produce a structurally valid, symmetric index."""
    from dblink_amd import ops

    if not ops.have_native():
        pytest.skip("native extension not built")
    # 300 distinct CJK-range characters + a few similar Latin names
    wide = ["".join(chr(0x4E00 + 7 * i + j) for j in range(3)) for i in range(300)]
    values = sorted(set(wide) | {"ANNA", "ANNE", "BOB"})
    fn = LevenshteinSimilarityFn(5.0, 10.0)
    idx = ops.sim_pairs(values, fn)
    V = len(values)
    assert idx.row_ptr.shape == (V + 1,)
    assert (np.diff(idx.row_ptr) >= 0).all()
    ia = values.index("ANNA")
    ib = values.index("ANNE")
    # ANNA ~ ANNE must be mutually similar whatever the byte encoding
    assert ib in idx.col[idx.row_ptr[ia]:idx.row_ptr[ia + 1]]
    assert ia in idx.col[idx.row_ptr[ib]:idx.row_ptr[ib + 1]]


def test_index_invariants_randomized():
    """Property test: normalizers equal the brute-force 1/sum phi*expsim and
    cached power distributions are exactly phi*norm^k normalized."""
    import math

    from hypothesis import given, settings
    from hypothesis import strategies as st

    @settings(max_examples=25, deadline=None)
    @given(
        values=st.lists(st.text(alphabet="ABC", min_size=0, max_size=8),
                        min_size=2, max_size=12, unique=True),
        weights=st.lists(st.floats(min_value=0.1, max_value=50.0),
                         min_size=2, max_size=12),
        threshold=st.floats(min_value=0.5, max_value=8.0),
    )
    def check(values, weights, threshold):
        k = min(len(values), len(weights))
        values, weights = sorted(values)[:k], weights[:k]
        if k < 2:
            return
        fn = LevenshteinSimilarityFn(threshold, 10.0)
        idx = AttributeIndex(dict(zip(values, weights)), fn, precache_powers=3,
                             pair_sweep=_python_sim_pairs)
        V = idx.num_values
        for v in range(V):
            tot = sum(idx.probs[w] * math.exp(fn.similarity(values[w], values[v]))
                      for w in range(V))
            assert abs(idx.sim_norms[v] - 1.0 / tot) < 1e-9 / tot + 1e-12
        for kpow in (1, 2, 3):
            d = idx.sim_norm_dist(kpow)
            w = idx.probs * idx.sim_norms ** kpow
            np.testing.assert_allclose(d.probs, w / w.sum(), rtol=1e-9)
            assert abs(idx.sim_norm_total(kpow) - w.sum()) < 1e-9 * w.sum()

    check()
