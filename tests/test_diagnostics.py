import numpy as np
import pytest

from dblink_amd.analysis.diagnostics import ess, summarize


def test_ess_iid():
    rng = np.random.default_rng(0)
    x = rng.normal(size=4000)
    e = ess(x)
    assert 2500 < e <= 4000  # iid: ESS ~ n


def test_ess_correlated():
    rng = np.random.default_rng(1)
    # AR(1) with phi = 0.9 -> tau ~ (1+phi)/(1-phi) = 19 -> ESS ~ n/19
    n = 8000
    x = np.empty(n)
    x[0] = 0
    for i in range(1, n):
        x[i] = 0.9 * x[i - 1] + rng.normal()
    e = ess(x)
    assert 150 < e < 1200, e


def test_ess_constant_and_tiny():
    assert ess(np.ones(100)) == 100
    assert ess(np.array([1.0, 2.0])) == 2


def test_summarize_from_run(tmp_path):
    # synthesize a diagnostics.csv
    p = tmp_path / "diagnostics.csv"
    rows = ["iteration,systemTime-ms,numObservedEntities,logLikelihood,popSize,aggDist-a,recDistortion-0,recDistortion-1"]
    rng = np.random.default_rng(2)
    t0 = 1_000_000
    for i in range(200):
        rows.append(f"{i},{t0 + i * 50},{90 + rng.integers(0, 5)},{-1000 + rng.normal():.3f},100,3,90,10")
    p.write_text("\n".join(rows) + "\n")
    out = summarize(str(tmp_path))
    assert out["iterations_per_sec"] == pytest.approx(20.0, rel=0.01)
    assert out["ess_logLikelihood"] > 50
    assert "ess_logLikelihood_per_sec" in out


def test_fast_smpc_matches_reference_impl(tmp_path):
    """Vectorized MPC/sMPC must agree with the direct Python implementation."""
    import pyarrow as pa

    from dblink_amd.analysis.chain import (
        most_probable_clusters,
        most_probable_clusters_fast,
        shared_most_probable_clusters,
        shared_most_probable_clusters_fast,
    )

    rng = np.random.default_rng(5)
    rows = {"iteration": [], "partitionId": [], "linkageStructure": []}
    rec = [f"r{i}" for i in range(30)]
    for it in range(12):
        # random clustering of 30 records into ~12 clusters, across 2 partitions
        labels = rng.integers(0, 12, 30)
        clusters = [[rec[i] for i in np.flatnonzero(labels == l)] for l in range(12)]
        clusters = [c for c in clusters if c]
        half = len(clusters) // 2
        for pid, cl in ((0, clusters[:half]), (1, clusters[half:])):
            rows["iteration"].append(it)
            rows["partitionId"].append(pid)
            rows["linkageStructure"].append(cl)
    table = pa.table(rows, schema=pa.schema([
        ("iteration", pa.int64()), ("partitionId", pa.int32()),
        ("linkageStructure", pa.list_(pa.list_(pa.string()))),
    ]))
    ref = most_probable_clusters(table)
    fast = most_probable_clusters_fast(table)
    assert set(ref) == set(fast)
    for rid in ref:
        assert ref[rid][0] == fast[rid][0], rid
        assert ref[rid][1] == pytest.approx(fast[rid][1])
    s_ref = sorted(map(sorted, shared_most_probable_clusters(table)))
    s_fast = sorted(map(sorted, shared_most_probable_clusters_fast(table)))
    assert s_ref == s_fast


def test_sampler_burnin_thinning_record_rule(tmp_path):
    """Exact recorded-iteration set under burn-in + thinning (the reference's
    ``>= burninInterval`` record rule, Sampler.scala:92-115): with burnin=5,
    thin=3, sampleSize=4 and iteration 0 also recorded only when burnin=0."""
    import os
    import sys

    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    import pyarrow.parquet  # noqa: F401
    from bench import build_cache_and_records
    from dblink_amd.analysis.chain import load_chain
    from dblink_amd.engine import sampler as sampler_m
    from dblink_amd.engine.cpu_engine import CpuEngine, SamplerFlags
    from dblink_amd.engine.init import deterministic_init
    from dblink_amd.parallel.partitioning import KDTreePartitioner

    n = 40
    cache, rv, rf = build_cache_and_records(n, seed=6)
    partitioner = KDTreePartitioner(0, [])
    state = deterministic_init(rv, rf, np.arange(n, dtype=np.int64), cache,
                               partitioner, seed=6)
    engine = CpuEngine(cache, partitioner)
    engine.rec_ids_array = np.array([str(i) for i in range(n)], dtype=object)
    engine.initial_summary(state)
    out = str(tmp_path / "o1")
    sampler_m.sample(engine, state, 4, out, burnin_interval=5,
                     thinning_interval=3, checkpoint_interval=0,
                     flags=SamplerFlags.for_sampler("PCG-I"))
    got = sorted(set(load_chain(out)["iteration"].to_pylist()))
    assert got == [5, 8, 11, 14], got

    # burnin = 0 records the INITIAL state too (iteration 0)
    state2 = deterministic_init(rv, rf, np.arange(n, dtype=np.int64), cache,
                                partitioner, seed=6)
    engine2 = CpuEngine(cache, partitioner)
    engine2.rec_ids_array = engine.rec_ids_array
    engine2.initial_summary(state2)
    out2 = str(tmp_path / "o2")
    sampler_m.sample(engine2, state2, 3, out2, burnin_interval=0,
                     thinning_interval=2, checkpoint_interval=0,
                     flags=SamplerFlags.for_sampler("PCG-I"))
    got2 = sorted(set(load_chain(out2)["iteration"].to_pylist()))
    assert got2 == [0, 2, 4, 6], got2


def test_mpc_fast_equals_slow_randomized():
    """Property test: the hash-aggregated MPC/sMPC equals the reference
    per-cluster implementation on random synthetic chains."""
    import pyarrow as pa
    from hypothesis import given, settings
    from hypothesis import strategies as st

    from dblink_amd.analysis.chain import (
        most_probable_clusters,
        most_probable_clusters_fast,
        shared_most_probable_clusters,
        shared_most_probable_clusters_fast,
    )
    from dblink_amd.engine.writers import LinkageChainWriter

    def make_table(memberships):
        rows = {"iteration": [], "partitionId": [], "linkageStructure": []}
        for it, m in enumerate(memberships):
            clusters = {}
            for rid, c in enumerate(m):
                clusters.setdefault(int(c), []).append(str(rid))
            rows["iteration"].append(it)
            rows["partitionId"].append(0)
            rows["linkageStructure"].append(list(clusters.values()))
        return pa.table(rows, schema=LinkageChainWriter.SCHEMA)

    @settings(max_examples=30, deadline=None)
    @given(st.integers(min_value=2, max_value=12),
           st.integers(min_value=1, max_value=8),
           st.integers(min_value=0, max_value=2**31))
    def check(n, iters, seed):
        rng = np.random.default_rng(seed)
        memberships = [rng.integers(0, max(2, n // 2), n) for _ in range(iters)]
        t = make_table(memberships)
        slow = most_probable_clusters(t)
        fast = most_probable_clusters_fast(t)
        assert set(slow) == set(fast)
        for rid in slow:
            assert slow[rid][0] == fast[rid][0]
            assert abs(slow[rid][1] - fast[rid][1]) < 1e-9
        assert ({frozenset(c) for c in shared_most_probable_clusters(t)}
                == {frozenset(c) for c in shared_most_probable_clusters_fast(t)})

    check()


def test_fast_mpc_lexsort_fallback(monkeypatch):
    """Chains too large for the packed int64 composite sort fall back to a
    3-key lexsort; force that branch and check it agrees with the direct
    implementation."""
    import pyarrow as pa

    from dblink_amd.analysis import chain as chain_mod
    from dblink_amd.analysis.chain import (
        most_probable_clusters,
        most_probable_clusters_fast,
        shared_most_probable_clusters,
        shared_most_probable_clusters_fast,
    )

    monkeypatch.setattr(chain_mod, "_PACK_BITS", 1)  # always too narrow
    rng = np.random.default_rng(9)
    rows = {"iteration": [], "partitionId": [], "linkageStructure": []}
    rec = [f"r{i}" for i in range(25)]
    for it in range(10):
        labels = rng.integers(0, 10, 25)
        clusters = [[rec[i] for i in np.flatnonzero(labels == l)]
                    for l in range(10)]
        rows["iteration"].append(it)
        rows["partitionId"].append(0)
        rows["linkageStructure"].append([c for c in clusters if c])
    table = pa.table(rows, schema=pa.schema([
        ("iteration", pa.int64()), ("partitionId", pa.int32()),
        ("linkageStructure", pa.list_(pa.list_(pa.string()))),
    ]))
    ref = most_probable_clusters(table)
    fast = most_probable_clusters_fast(table)
    assert set(ref) == set(fast)
    for rid in ref:
        assert ref[rid][0] == fast[rid][0], rid
        assert ref[rid][1] == pytest.approx(fast[rid][1])
    assert (sorted(map(sorted, shared_most_probable_clusters(table)))
            == sorted(map(sorted, shared_most_probable_clusters_fast(table))))


def test_fast_mpc_matches_reference_fuzz():
    """Property fuzz: on random linkage chains (random cluster counts,
    sizes, partition splits, missing records per sample) the vectorized
    MPC/sMPC must agree with the direct python implementation."""
    import pyarrow as pa
    from hypothesis import given, settings
    from hypothesis import strategies as st

    from dblink_amd.analysis.chain import (
        most_probable_clusters,
        most_probable_clusters_fast,
        shared_most_probable_clusters,
        shared_most_probable_clusters_fast,
    )

    @settings(max_examples=40, deadline=None)
    @given(st.integers(0, 2**31 - 1), st.integers(4, 25), st.integers(2, 8))
    def check(seed, n_rec, n_samples):
        rng = np.random.default_rng(seed)
        rec = [f"r{i}" for i in range(n_rec)]
        rows = {"iteration": [], "partitionId": [], "linkageStructure": []}
        for it in range(n_samples):
            # random subset of records present (a record can be absent from
            # a partition's sample), random clustering, random 2-way split
            present = np.flatnonzero(rng.random(n_rec) < 0.9)
            labels = rng.integers(0, max(1, n_rec // 2), len(present))
            clusters = {}
            for r, l in zip(present, labels):
                clusters.setdefault(int(l), []).append(rec[r])
            cl = list(clusters.values())
            half = rng.integers(0, len(cl) + 1)
            for pid, part in ((0, cl[:half]), (1, cl[half:])):
                rows["iteration"].append(it)
                rows["partitionId"].append(pid)
                rows["linkageStructure"].append(part)
        table = pa.table(rows, schema=pa.schema([
            ("iteration", pa.int64()), ("partitionId", pa.int32()),
            ("linkageStructure", pa.list_(pa.list_(pa.string()))),
        ]))
        ref = most_probable_clusters(table)
        fast = most_probable_clusters_fast(table)
        assert set(ref) == set(fast)
        for rid in ref:
            assert ref[rid][0] == fast[rid][0], rid
            assert abs(ref[rid][1] - fast[rid][1]) < 1e-12
        assert (sorted(map(sorted, shared_most_probable_clusters(table)))
                == sorted(map(sorted, shared_most_probable_clusters_fast(table))))

    check()
