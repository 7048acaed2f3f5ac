"""Reproduce the CPU-engine stationary-sweep measurements in BENCH.md on
the reference's real RLdata10000 dataset (verbatim attribute/prior setup):
burn 300 sweeps, time 300, optional cProfile breakdown with --profile."""
import sys, time
sys.path.insert(0, "/root/repo")
import numpy as np

RLDATA = "/root/reference/examples/RLdata10000.csv"


def build():
    import csv

    from dblink_amd.engine.cpu_engine import CpuEngine
    from dblink_amd.engine.init import deterministic_init
    from dblink_amd.models.records import (Attribute, BetaShapeParameters,
                                           RecordsCache, RecordsTable)
    from dblink_amd.models.similarity import (ConstantSimilarityFn,
                                              LevenshteinSimilarityFn)
    from dblink_amd.parallel.partitioning import KDTreePartitioner

    rows = list(csv.DictReader(open(RLDATA)))
    cols = ["fname_c1", "lname_c1", "by", "bm", "bd"]
    table = RecordsTable.from_rows(
        [r["rec_id"] for r in rows], ["0"] * len(rows),
        [[r[c] if r[c] != "NA" else None for c in cols] for r in rows])
    prior = BetaShapeParameters(10.0, 1000.0)
    attrs = [Attribute("fname_c1", LevenshteinSimilarityFn(7.0, 10.0), prior),
             Attribute("lname_c1", LevenshteinSimilarityFn(7.0, 10.0), prior),
             Attribute("by", ConstantSimilarityFn(), prior),
             Attribute("bm", ConstantSimilarityFn(), prior),
             Attribute("bd", ConstantSimilarityFn(), prior)]
    cache = RecordsCache.build(table, attrs, max_cluster_size=10)
    rv, rf = cache.transform_records(table)
    part = KDTreePartitioner(3, [0, 1])
    state = deterministic_init(rv, rf, np.arange(len(rows), dtype=np.int64),
                               cache, part, seed=1)
    engine = CpuEngine(cache, part)
    engine.initial_summary(state)
    return engine, state


def main():
    from dblink_amd.engine.cpu_engine import SamplerFlags

    engine, state = build()
    flags = SamplerFlags.for_sampler("PCG-I")
    for _ in range(300):
        engine.step(state, flags)
    t0 = time.time()
    for _ in range(300):
        engine.step(state, flags)
    ms = (time.time() - t0) / 300 * 1e3
    print(f"real RLdata10000 stationary: {ms:.2f} ms/sweep "
          f"loglik={state.summary.log_likelihood:.6f}")
    if "--profile" in sys.argv:
        import cProfile, io, pstats

        pr = cProfile.Profile()
        pr.enable()
        for _ in range(300):
            engine.step(state, flags)
        pr.disable()
        s = io.StringIO()
        pstats.Stats(pr, stream=s).sort_stats("tottime").print_stats(15)
        print("\n".join(s.getvalue().splitlines()[4:25]))


if __name__ == "__main__":
    main()
