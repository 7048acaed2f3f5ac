"""Alias-sampler statistical tests (parity with AliasSamplerTest.scala:25-73,
with a reduced draw count for CI speed)."""

import numpy as np
import pytest

from dblink_amd.models.alias import AliasTable


def test_invalid_weights():
    with pytest.raises(ValueError):
        AliasTable(np.array([1.0, -0.5]))
    with pytest.raises(ValueError):
        AliasTable(np.array([np.nan, 1.0]))
    with pytest.raises(ValueError):
        AliasTable(np.array([np.inf, 1.0]))
    with pytest.raises(ValueError):
        AliasTable(np.array([0.0, 0.0]))
    with pytest.raises(ValueError):
        AliasTable(np.array([]))


def test_empirical_distribution_matches():
    rng = np.random.default_rng(1)
    weights = np.array([0.2, 0.3, 0.1, 0.4])
    table = AliasTable(weights)
    draws = table.sample(rng, size=1_000_000)
    emp = np.bincount(draws, minlength=4) / 1_000_000
    np.testing.assert_allclose(emp, weights, atol=2e-3)


def test_extreme_distribution():
    # AliasSamplerTest.scala:58-72: [1e-9, 1e-9, ~1]
    rng = np.random.default_rng(1)
    w = np.array([1e-9, 1e-9, 1.0 - 2e-9])
    table = AliasTable(w)
    draws = table.sample(rng, size=100_000)
    assert np.all(draws == 2) or np.mean(draws == 2) > 0.9999


def test_zero_probability_never_sampled():
    rng = np.random.default_rng(7)
    table = AliasTable(np.array([0.0, 1.0, 0.0, 2.0]))
    draws = table.sample(rng, size=10_000)
    assert set(np.unique(draws)) <= {1, 3}


def test_unnormalized_weights():
    rng = np.random.default_rng(3)
    table = AliasTable(np.array([2.0, 6.0]))
    draws = table.sample(rng, size=200_000)
    assert np.mean(draws == 1) == pytest.approx(0.75, abs=5e-3)
