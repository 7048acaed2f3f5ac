"""A/B the chunked hash-accumulate dense value path (DBLINK_VCHUNK) vs the
union-merge path, at the 1M-64 and 10M-16 stationary configs (one GPU).
DBLINK_SIMH stays ON in both arms (it is orthogonal and already validated).
"""
import os, sys, time
sys.path.insert(0, "/root/repo")
import numpy as np
import torch

import bench as b
from dblink_amd.engine.cpu_engine import SamplerFlags
from dblink_amd.engine.gpu_engine import GpuEngine
from dblink_amd.engine.init import deterministic_init
from dblink_amd.parallel.partitioning import KDTreePartitioner

dev = torch.device("cuda", 0)
flags = SamplerFlags.for_sampler("PCG-I")


def build(n, levels, schema="rldata", seed=77):
    cache, rv, rf = b.build_cache_and_records(n, seed=seed, schema=schema)
    part_attrs = [3, 4] if schema == "rldata" else [0, 1]
    part = KDTreePartitioner(levels, part_attrs)
    state = deterministic_init(rv, rf, np.arange(n, dtype=np.int64), cache,
                               part, seed=319158)
    engine = GpuEngine(cache, part, device=dev)
    engine.initial_summary(state)
    return engine, state


def time_arm(engine, state, sweeps, vchunk):
    os.environ["DBLINK_VCHUNK"] = str(vchunk)
    for _ in range(10):
        engine.step(state, flags)
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(sweeps):
        engine.step(state, flags)
    torch.cuda.synchronize()
    return (time.time() - t0) / sweeps * 1e3


def ab(tag, n, levels, schema, burn, sweeps):
    t0 = time.time()
    engine, state = build(n, levels, schema)
    print(f"[ab] {tag} init {time.time()-t0:.1f}s", flush=True)
    t0 = time.time()
    for _ in range(burn):
        engine.step(state, flags)
    torch.cuda.synchronize()
    print(f"[ab] {tag} burn {burn} in {time.time()-t0:.1f}s", flush=True)
    # interleaved ABBA blocks cancel the slow drift of a still-evolving chain
    acc = {0: [], 1: []}
    for arm in (1, 0, 0, 1, 1, 0, 0, 1):
        acc[arm].append(time_arm(engine, state, sweeps, arm))
    m1 = np.mean(acc[1]); m0 = np.mean(acc[0])
    print(f"[ab] {tag} ms/sweep vchunk=1: {m1:.2f} {['%.1f' % x for x in acc[1]]}  "
          f"vchunk=0: {m0:.2f} {['%.1f' % x for x in acc[0]]}", flush=True)


ab("10M-16", 10_000_000, 4, "strings8", 400, 20)
ab("1M-64", 1_000_000, 6, "rldata", 300, 40)
