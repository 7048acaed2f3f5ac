"""Deep 10M-16 soak: per-100-sweep block cost until the time budget runs
out, to locate (or bound) the stationary plateau BENCH.md's config #4 note
discusses."""
import sys, time
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
t0 = time.time()
cache, rv, rf = b.build_cache_and_records(10_000_000, seed=77, schema="strings8")
part = KDTreePartitioner(4, [0, 1])
state = deterministic_init(rv, rf, np.arange(10_000_000, dtype=np.int64),
                           cache, part, seed=319158)
engine = GpuEngine(cache, part, device=dev)
engine.initial_summary(state)
print(f"[soak] init {time.time()-t0:.1f}s", flush=True)

deadline = time.time() + 280
sweep = 0
while time.time() < deadline:
    t0 = time.time()
    for _ in range(100):
        engine.step(state, flags)
    torch.cuda.synchronize()
    sweep += 100
    print(f"[soak] sweeps {sweep - 100}-{sweep}: "
          f"{(time.time() - t0) * 10:.1f} ms/sweep", flush=True)
