"""Multi-process (gloo, world_size=2) tests of the distributed path:
migration all-to-all, summary all-reduce, and a short end-to-end chain.

These exercise the SAME code path the 8-GPU RCCL run uses (comm.py switches
backend only), per the reference's pseudocluster testing strategy
(SURVEY.md §4)."""

import json
import os
import subprocess
import sys

import numpy as np
import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

WORKER = r"""
import json, os, sys
sys.path.insert(0, "__ROOT__")
import numpy as np
import torch.distributed as dist

from dblink_amd.parallel import comm
from dblink_amd.parallel.partitioning import KDTreePartitioner
from dblink_amd.engine.init import deterministic_init
from dblink_amd.engine.cpu_engine import CpuEngine, SamplerFlags
from bench import build_cache_and_records

rank, world, device = comm.init_from_env(backend="gloo")

n = 240
cache, rec_values, rec_files = build_cache_and_records(n, seed=11)
partitioner = KDTreePartitioner(__LEVELS__, [3, 4, 0][:__LEVELS__])
bounds = np.linspace(0, n, world + 1).astype(np.int64)
lo, hi = int(bounds[rank]), int(bounds[rank + 1])
state = deterministic_init(rec_values[lo:hi], rec_files[lo:hi],
                           np.arange(lo, hi, dtype=np.int64), cache, partitioner,
                           seed=5, rank=rank, world_size=world)

# ownership invariant after init migration: pid % world == rank
assert np.all(state.ent_part % world == rank), "entity placed on wrong rank"

engine = CpuEngine(cache, partitioner, world_size=world, rank=rank)
engine.initial_summary(state)
flags = SamplerFlags.for_sampler(os.environ.get("DBLINK_TEST_SAMPLER", "PCG-I"))

lls = []
for i in range(25):
    engine.step(state, flags)
    assert np.all(state.ent_part % world == rank)
    # rec_ent indices valid
    assert state.rec_ent.min() >= 0 and state.rec_ent.max() < state.num_entities
    lls.append(state.summary.log_likelihood)

# conservation: global entity and record counts unchanged
import torch
t = torch.tensor([state.num_entities, state.num_records], dtype=torch.float64)
comm.all_reduce_sum_(t)
assert int(t[0]) == state.population_size, (int(t[0]), state.population_size)
assert int(t[1]) == n

# record gids form a partition of 0..n-1 globally
gids = comm.all_gather_object(sorted(state.rec_gid.tolist()))
if rank == 0:
    allg = sorted(g for lst in gids for g in lst)
    assert allg == list(range(n)), "records lost or duplicated in migration"
    print(json.dumps({"ok": True, "ll": lls[-1], "isolates": state.summary.num_isolates}))
dist.destroy_process_group()
"""


def _run_workers(script, world=2, timeout=600):
    env = dict(os.environ)
    env.update(
        MASTER_ADDR="127.0.0.1",
        MASTER_PORT=str(29500 + os.getpid() % 1000),
        WORLD_SIZE=str(world),
        GLOO_SOCKET_IFNAME="lo",
    )
    procs = []
    for r in range(world):
        e = dict(env, RANK=str(r), LOCAL_RANK=str(r))
        procs.append(
            subprocess.Popen(
                [sys.executable, "-c", script], env=e,
                stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True,
            )
        )
    outs = []
    for p in procs:
        out, _ = p.communicate(timeout=timeout)
        outs.append(out)
    for p, out in zip(procs, outs):
        assert p.returncode == 0, f"worker failed:\n{out}"
    return outs


@pytest.mark.slow
def test_two_rank_chain_conservation():
    outs = _run_workers(WORKER.replace("__ROOT__", ROOT).replace("__LEVELS__", "2"))
    payload = None
    for out in outs:
        for line in out.splitlines():
            if line.startswith("{"):
                payload = json.loads(line)
    assert payload and payload["ok"]
    assert np.isfinite(payload["ll"])


SINGLE_VS_TWO = r"""
import json, os, sys
sys.path.insert(0, "__ROOT__")
import numpy as np
import torch.distributed as dist

from dblink_amd.parallel import comm
from dblink_amd.parallel.partitioning import KDTreePartitioner
from dblink_amd.engine.init import deterministic_init
from dblink_amd.engine.cpu_engine import CpuEngine, SamplerFlags
from bench import build_cache_and_records

rank, world, device = comm.init_from_env(backend="gloo")
n = 160
cache, rec_values, rec_files = build_cache_and_records(n, seed=3)
partitioner = KDTreePartitioner(1, [3])
bounds = np.linspace(0, n, world + 1).astype(np.int64)
lo, hi = int(bounds[rank]), int(bounds[rank + 1])
state = deterministic_init(rec_values[lo:hi], rec_files[lo:hi],
                           np.arange(lo, hi, dtype=np.int64), cache, partitioner,
                           seed=2, rank=rank, world_size=world)
engine = CpuEngine(cache, partitioner, world_size=world, rank=rank)
engine.initial_summary(state)
flags = SamplerFlags.for_sampler("PCG-I")
tail = []
for i in range(60):
    engine.step(state, flags)
    if i >= 40:
        tail.append(state.summary.log_likelihood)
if rank == 0:
    print(json.dumps({"mean_ll": float(np.mean(tail))}))
dist.destroy_process_group()
"""


@pytest.mark.slow
@pytest.mark.slow
@pytest.mark.parametrize("sampler", ["PCG-II", "Gibbs"])
def test_two_rank_variants_conservation(sampler):
    """The dense (PCG-II) and plain-Gibbs fast paths under real
    multi-process migration: same conservation/ownership invariants."""
    import os as _os

    _os.environ["DBLINK_TEST_SAMPLER"] = sampler
    try:
        outs = _run_workers(WORKER.replace("__ROOT__", ROOT).replace("__LEVELS__", "2"))
        assert any('"ok": true' in o for o in outs), outs
    finally:
        _os.environ.pop("DBLINK_TEST_SAMPLER", None)


@pytest.mark.slow
def test_four_rank_chain_conservation():
    """Four ranks over eight KD partitions: ownership, conservation and gid
    coverage invariants at a world size with multi-partition-per-rank
    migration fan-out (same path as the 8-GPU run)."""
    outs = _run_workers(WORKER.replace("__ROOT__", ROOT).replace("__LEVELS__", "3"),
                        world=4)
    assert any('"ok": true' in o for o in outs), outs


def test_two_rank_posterior_matches_single_rank():
    """The 2-rank chain must land in the same stationary log-likelihood band
    as the 1-rank chain on identical data (same model, different RNG streams)."""
    # single rank, in-process
    sys.path.insert(0, ROOT)
    from bench import build_cache_and_records
    from dblink_amd.engine.cpu_engine import CpuEngine, SamplerFlags
    from dblink_amd.engine.init import deterministic_init
    from dblink_amd.parallel.partitioning import KDTreePartitioner

    n = 160
    cache, rec_values, rec_files = build_cache_and_records(n, seed=3)
    partitioner = KDTreePartitioner(1, [3])
    state = deterministic_init(rec_values, rec_files, np.arange(n, dtype=np.int64),
                               cache, partitioner, seed=2)
    engine = CpuEngine(cache, partitioner)
    engine.initial_summary(state)
    flags = SamplerFlags.for_sampler("PCG-I")
    tail = []
    for i in range(60):
        engine.step(state, flags)
        if i >= 40:
            tail.append(state.summary.log_likelihood)
    single = float(np.mean(tail))

    outs = _run_workers(SINGLE_VS_TWO.replace("__ROOT__", ROOT))
    two = None
    for out in outs:
        for line in out.splitlines():
            if line.startswith("{"):
                two = json.loads(line)["mean_ll"]
    assert two is not None
    assert abs(single - two) / abs(single) < 0.05, (single, two)


TENSOR_MIGRATE = r"""
import json, os, sys
sys.path.insert(0, "__ROOT__")
import numpy as np
import torch
import torch.distributed as dist
from types import SimpleNamespace

from dblink_amd.parallel import comm
from dblink_amd.parallel.migration import migrate_and_sort_tensors

rank, world, device = comm.init_from_env(backend="gloo")
rng = np.random.default_rng(100 + rank)
E, R, A, P = 50, 80, 3, 8
gs = SimpleNamespace(
    ent_values=torch.tensor(rng.integers(0, 9, (E, A)), dtype=torch.int32),
    ent_part=torch.tensor(rng.integers(0, P, E), dtype=torch.int32),
    rec_values=torch.tensor(rng.integers(-1, 9, (R, A)), dtype=torch.int32),
    rec_file=torch.zeros(R, dtype=torch.int32),
    rec_dist=torch.tensor(rng.integers(0, 2, (R, A)), dtype=torch.uint8),
    rec_gid=torch.tensor(rank * R + np.arange(R), dtype=torch.int64),
    rec_ent=torch.tensor(rng.integers(0, E, R), dtype=torch.int64),
    rec_part=None,
)
# remember each record's entity VALUES so we can verify links survive the move
before = {int(g): gs.ent_values[gs.rec_ent[i]].tolist() for i, g in enumerate(gs.rec_gid)}
allbefore = comm.all_gather_object(before)
merged = {}
for d in allbefore:
    merged.update(d)

migrate_and_sort_tensors(gs, world)

# ownership + sortedness invariants
assert torch.all(gs.ent_part.to(torch.int64) % world == rank)
assert torch.all(gs.ent_part[1:] >= gs.ent_part[:-1])
assert torch.all(gs.rec_ent[1:] >= gs.rec_ent[:-1])
assert torch.all(gs.rec_part == gs.ent_part[gs.rec_ent])
# every record still points at an entity with its original values
for i in range(gs.rec_gid.numel()):
    g = int(gs.rec_gid[i])
    assert gs.ent_values[gs.rec_ent[i]].tolist() == merged[g], g
counts = comm.all_gather_object((gs.ent_values.shape[0], gs.rec_gid.numel()))
if rank == 0:
    assert sum(c[0] for c in counts) == world * E
    assert sum(c[1] for c in counts) == world * R
    print(json.dumps({"ok": True}))
dist.destroy_process_group()
"""


@pytest.mark.slow
def test_tensor_migration_preserves_links():
    """migrate_and_sort_tensors (the GPU engine's migration) must preserve
    every record->entity link across the all-to-all (gloo, CPU tensors)."""
    outs = _run_workers(TENSOR_MIGRATE.replace("__ROOT__", ROOT))
    assert any('"ok": true' in o for o in outs), outs


ASYNC_OVERLAP = r"""
import json, sys
sys.path.insert(0, "__ROOT__")
import torch
import torch.distributed as dist
from dblink_amd.parallel import comm

rank, world, device = comm.init_from_env(backend="gloo")
# start async summary reduce on the second group, then run an all_to_all on
# the default group before waiting (the GPU engine's overlap pattern)
t = torch.full((8,), float(rank + 1), dtype=torch.float64)
work = comm.all_reduce_sum_async(t)
payload = torch.full((4, 3), rank, dtype=torch.int32)
out, counts = comm.all_to_all_v(payload, [2] * world)
work.wait()
expect = sum(range(1, world + 1))
assert torch.all(t == expect), t
assert out.shape[0] == 2 * world
if rank == 0:
    print(json.dumps({"ok": True}))
dist.destroy_process_group()
"""


@pytest.mark.slow
def test_async_summary_reduce_overlaps_alltoall():
    outs = _run_workers(ASYNC_OVERLAP.replace("__ROOT__", ROOT))
    assert any('"ok": true' in o for o in outs), outs


PROJECT_WORKER = r"""
import sys
sys.path.insert(0, "__ROOT__")
from dblink_amd.api.cli import main
raise SystemExit(main(["__CONF__"]))
"""


@pytest.mark.slow
def test_multi_rank_project_resume(tmp_path):
    """Full project flow at world 2 through the CLI twice (fresh + resume):
    per-rank parquet parts, partition structure and record coverage survive
    the resume (exercises the saved-partitioner restore at world > 1)."""
    import sys as _sys

    _sys.path.insert(0, ROOT)
    from dblink_amd.analysis.chain import load_chain
    from dblink_amd.utils.synthdata import write_csv

    data = str(tmp_path / "d.csv")
    write_csv(data, 300, dup_fraction=0.1, seed=4)
    out = str(tmp_path / "res")
    conf = tmp_path / "p.conf"
    conf_text = """
    dblink : {
      lowDistortion : {alpha : 0.5, beta : 50.0}
      data : { path : "%s", recordIdentifier : "rec_id",
               entityIdentifier : "ent_id", nullValue : "NA",
               matchingAttributes : [
        {name : "by", similarityFunction : {name : "ConstantSimilarityFn"}, distortionPrior : ${dblink.lowDistortion}},
        {name : "bm", similarityFunction : {name : "ConstantSimilarityFn"}, distortionPrior : ${dblink.lowDistortion}},
        {name : "fname_c1", similarityFunction : {name : "LevenshteinSimilarityFn", parameters : {threshold : 7.0, maxSimilarity : 10.0}}, distortionPrior : ${dblink.lowDistortion}},
        {name : "lname_c1", similarityFunction : {name : "LevenshteinSimilarityFn", parameters : {threshold : 7.0, maxSimilarity : 10.0}}, distortionPrior : ${dblink.lowDistortion}} ] }
      randomSeed : 7
      engine : "cpu"
      partitioner : {name : "KDTreePartitioner",
                     parameters : {numLevels : 2, matchingAttributes : ["fname_c1", "lname_c1"]}}
      outputPath : "%s/"
      checkpointPath : "%s/ckpt/"
      steps : [{name : "sample", parameters : {
        sampleSize : 6, burninInterval : 4, thinningInterval : 2,
        resume : %s, sampler : "PCG-I", checkpointInterval : 0}}]
    }
    """
    script = PROJECT_WORKER.replace("__ROOT__", ROOT).replace("__CONF__", str(conf))
    conf.write_text(conf_text % (data, out, out, "false"))
    _run_workers(script, world=2)
    first = max(load_chain(out)["iteration"].to_pylist())
    conf.write_text(conf_text % (data, out, out, "true"))
    _run_workers(script, world=2)
    t = load_chain(out)
    assert max(t["iteration"].to_pylist()) > first
    assert sorted(set(t["partitionId"].to_pylist())) == [0, 1, 2, 3]
    import collections

    per_iter = collections.Counter()
    for row in t.to_pylist():
        per_iter[row["iteration"]] += sum(len(c) for c in row["linkageStructure"])
    assert set(per_iter.values()) == {300}


OVERLAP_MIGRATE = r"""
import copy, json, os, sys
sys.path.insert(0, "__ROOT__")
import numpy as np
import torch
import torch.distributed as dist
from types import SimpleNamespace

from dblink_amd.parallel import comm
from dblink_amd.parallel.migration import migrate_and_sort_tensors, migrate_overlapped

rank, world, device = comm.init_from_env(backend="gloo")
rng = np.random.default_rng(100 + rank)
E, R, A, P = 60, 100, 4, 2 * world

def make_gs():
    return SimpleNamespace(
        ent_values=torch.tensor(rng.integers(0, 9, (E, A)), dtype=torch.int32),
        ent_part=torch.tensor(rng.integers(0, P, E), dtype=torch.int32),
        rec_values=torch.tensor(rng.integers(-1, 9, (R, A)), dtype=torch.int32),
        rec_file=torch.zeros(R, dtype=torch.int32),
        rec_dist=torch.tensor(rng.integers(0, 2, (R, A)), dtype=torch.uint8),
        rec_gid=torch.tensor(rank * R + np.arange(R), dtype=torch.int64),
        rec_ent=torch.tensor(rng.integers(0, E, R), dtype=torch.int64),
        rec_part=None,
    )

FIELDS = ["ent_values", "ent_part", "rec_values", "rec_file", "rec_dist",
          "rec_gid", "rec_ent", "rec_part"]

rounds = int(os.environ.get("DBLINK_TEST_ROUNDS", "6"))
flight_ran = 0
for rnd in range(rounds):
    gs_a = make_gs()
    gs_b = SimpleNamespace(**{f: getattr(gs_a, f).clone()
                              for f in FIELDS if getattr(gs_a, f) is not None},
                           rec_part=None)
    migrate_and_sort_tensors(gs_a, world)

    def flight():
        global flight_ran
        flight_ran += 1

    migrate_overlapped(gs_b, world, rank, during_flight=flight)
    # bitwise parity with the eager reference path
    for f in FIELDS:
        ta, tb = getattr(gs_a, f), getattr(gs_b, f)
        assert torch.equal(ta, tb), (rnd, f, ta, tb)
    # invariants
    assert torch.all(gs_b.ent_part.to(torch.int64) % world == rank)
    assert torch.all(gs_b.ent_part[1:] >= gs_b.ent_part[:-1])
    assert torch.all(gs_b.rec_ent[1:] >= gs_b.rec_ent[:-1])
    # global conservation of entities/records and exact gid coverage
    t = torch.tensor([gs_b.ent_values.shape[0], gs_b.rec_gid.numel()],
                     dtype=torch.float64)
    comm.all_reduce_sum_(t)
    assert int(t[0]) == world * E and int(t[1]) == world * R
    gids = comm.all_gather_object(sorted(gs_b.rec_gid.tolist()))
    if rank == 0:
        allg = sorted(g for lst in gids for g in lst)
        assert allg == list(range(world * R)), "records lost or duplicated"

assert flight_ran == rounds
if rank == 0:
    print(json.dumps({"ok": True, "rounds": rounds}))
dist.destroy_process_group()
"""


@pytest.mark.slow
def test_overlapped_migration_bitwise_matches_eager():
    """migrate_overlapped (async migrants-only path) must produce the exact
    post-state of migrate_and_sort_tensors, round after round (gloo, world 2)."""
    outs = _run_workers(OVERLAP_MIGRATE.replace("__ROOT__", ROOT))
    assert any('"ok": true' in o for o in outs), outs


@pytest.mark.slow
def test_overlapped_migration_eight_rank_soak():
    """8-rank soak of the overlapped migration: bitwise parity with the eager
    path plus conservation/ownership/gid-coverage invariants every round —
    the fan-out shape of the 8-GPU node."""
    import os as _os

    _os.environ["DBLINK_TEST_ROUNDS"] = "5"
    try:
        outs = _run_workers(OVERLAP_MIGRATE.replace("__ROOT__", ROOT), world=8,
                            timeout=900)
        assert any('"ok": true' in o for o in outs), outs
    finally:
        _os.environ.pop("DBLINK_TEST_ROUNDS", None)


@pytest.mark.slow
def test_eight_rank_chain_soak():
    """Eight ranks over sixteen KD partitions — the 8-GPU node's fan-out
    shape: ownership, conservation and gid-coverage invariants held across
    25 full sweeps with per-sweep migration (gloo)."""
    outs = _run_workers(WORKER.replace("__ROOT__", ROOT).replace("__LEVELS__", "4"),
                        world=8, timeout=900)
    assert any('"ok": true' in o for o in outs), outs
