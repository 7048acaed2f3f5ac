"""Golden-value similarity tests, constants cross-checked against the
reference's SimilarityFnTest.scala:25-75."""

import pytest

from dblink_amd.models.similarity import (
    ConstantSimilarityFn,
    LevenshteinSimilarityFn,
    levenshtein,
)


def test_levenshtein_distance():
    assert levenshtein("", "") == 0
    assert levenshtein("abc", "") == 3
    assert levenshtein("", "abc") == 3
    assert levenshtein("kitten", "sitting") == 3
    assert levenshtein("AB", "BB") == 1
    assert levenshtein("flaw", "lawn") == 2


def test_constant_similarity():
    fn = ConstantSimilarityFn()
    assert fn.similarity("a", "b") == 0.0
    assert fn.similarity("same", "same") == 0.0
    assert fn.is_constant


def test_levenshtein_identity_and_symmetry():
    fn = LevenshteinSimilarityFn(threshold=5.0, maxSimilarity=10.0) if False else LevenshteinSimilarityFn(5.0, 10.0)
    assert fn.similarity("hello", "hello") == pytest.approx(10.0)
    assert fn.similarity("abc", "xyz") == fn.similarity("xyz", "abc")


def test_levenshtein_golden_values():
    # Reference golden: "AB"/"BB" -> 2.0 with threshold 5, 6.0 with threshold 0
    with_thresh = LevenshteinSimilarityFn(5.0, 10.0)
    assert with_thresh.similarity("AB", "BB") == pytest.approx(2.0)
    no_thresh = LevenshteinSimilarityFn(0.0, 10.0)
    assert no_thresh.similarity("AB", "BB") == pytest.approx(6.0)


def test_threshold_truncation():
    fn = LevenshteinSimilarityFn(7.0, 10.0)
    # unit similarity must exceed 0.7 for non-zero similarity
    assert fn.similarity("abcdefgh", "zyxwvuts") == 0.0
    assert fn.similarity("abcdefgh", "abcdefgh") == pytest.approx(10.0)


def test_invalid_params():
    with pytest.raises(ValueError):
        LevenshteinSimilarityFn(10.0, 10.0)  # threshold must be < maxSimilarity
    with pytest.raises(ValueError):
        LevenshteinSimilarityFn(1.0, -1.0)


def test_empty_strings_unit_similarity():
    fn = LevenshteinSimilarityFn(5.0, 10.0)
    assert fn.unit_similarity("", "") == 1.0
    assert fn.similarity("", "") == pytest.approx(10.0)
