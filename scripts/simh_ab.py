"""A/B the LDS staged-sim-row path (DBLINK_SIMH) in the link kernels.

Three scales on one GPU:
  1. 10k/4part  — bitwise chain equality between arms (link_update_kernel)
  2. 100k/8part — bitwise chain equality (heavy kernel active) + arm timing
  3. 1M/64part  — burn to stationarity once, then time both arms

Both arms must be bitwise identical: the staged hash returns the exact same
float sims as the global binary search.
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


def build(n, levels, seed=77):
    cache, rv, rf = b.build_cache_and_records(n, seed=seed)
    part = KDTreePartitioner(levels, [3, 4])
    state = deterministic_init(rv, rf, np.arange(n, dtype=np.int64), cache,
                               part, seed=319158)
    engine = GpuEngine(cache, part, device=dev)
    engine.initial_summary(state)
    return engine, state


def run_chain(n, levels, sweeps, simh):
    os.environ["DBLINK_SIMH"] = str(simh)
    engine, state = build(n, levels)
    for _ in range(sweeps):
        engine.step(state, flags)
    torch.cuda.synchronize()
    gs = engine._gs
    return engine, state, gs.rec_ent.cpu().numpy().copy(), \
        gs.ent_values.cpu().numpy().copy()


def time_arm(engine, state, sweeps, simh):
    os.environ["DBLINK_SIMH"] = str(simh)
    for _ in range(10):
        engine.step(state, flags)
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(sweeps):
        engine.step(state, flags)
    torch.cuda.synchronize()
    return (time.time() - t0) / sweeps * 1e3


# ---- 1. 10k bitwise ----
_, _, re1, ev1 = run_chain(10_000, 2, 50, 1)
e0, s0, re0, ev0 = run_chain(10_000, 2, 50, 0)
assert (re1 == re0).all() and (ev1 == ev0).all(), "10k arms diverged"
print("[ab] 10k bitwise OK", flush=True)
a1 = time_arm(e0, s0, 400, 1)
a0 = time_arm(e0, s0, 400, 0)
print(f"[ab] 10k  ms/sweep simh=1: {a1:.3f}  simh=0: {a0:.3f}", flush=True)

# ---- 2. 100k bitwise (heavy active) ----
eng, st, re1, ev1 = run_chain(100_000, 3, 150, 1)
hs = eng._heavy_stats.cpu().numpy()
e0, s0, re0, ev0 = run_chain(100_000, 3, 150, 0)
assert (re1 == re0).all() and (ev1 == ev0).all(), "100k arms diverged"
print(f"[ab] 100k bitwise OK (heavy stats {hs}; heavy records must be > 0)",
      flush=True)
a1 = time_arm(e0, s0, 150, 1)
a0 = time_arm(e0, s0, 150, 0)
print(f"[ab] 100k ms/sweep simh=1: {a1:.3f}  simh=0: {a0:.3f}", flush=True)

# ---- 3. 1M-64 stationary timing ----
os.environ["DBLINK_SIMH"] = "1"
t0 = time.time()
engine, state = build(1_000_000, 6)
print(f"[ab] 1M init {time.time()-t0:.1f}s", flush=True)
t0 = time.time()
for _ in range(300):
    engine.step(state, flags)
torch.cuda.synchronize()
print(f"[ab] 1M burn 300 in {time.time()-t0:.1f}s", flush=True)
a1 = time_arm(engine, state, 120, 1)
a0 = time_arm(engine, state, 120, 0)
a1b = time_arm(engine, state, 120, 1)
print(f"[ab] 1M   ms/sweep simh=1: {a1:.2f}  simh=0: {a0:.2f}  "
      f"simh=1 again: {a1b:.2f}", flush=True)
