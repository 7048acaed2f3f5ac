"""Re-tune DBLINK_HEAVY_THRESH (exact-scan vs A* routing) at the 1M-64
stationary config after the LDS sim-row staging changed the per-candidate
cost of the exact path. ABA-style: the default is re-timed at both ends to
confirm the chain is flat."""
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

cache, rv, rf = b.build_cache_and_records(1_000_000, seed=77)
part = KDTreePartitioner(6, [3, 4])
state = deterministic_init(rv, rf, np.arange(1_000_000, dtype=np.int64),
                           cache, part, seed=319158)
engine = GpuEngine(cache, part, device=dev)
engine.initial_summary(state)
for _ in range(300):
    engine.step(state, flags)
torch.cuda.synchronize()
print("[sweep] burn done", flush=True)


def arm(th, sweeps=60):
    engine._heavy_thresh = th
    for _ in range(10):
        engine.step(state, flags)
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(sweeps):
        engine.step(state, flags)
    torch.cuda.synchronize()
    return (time.time() - t0) / sweeps * 1e3


for th in (2048, 1024, 4096, 8192, 16384, 512, 2048):
    print(f"[sweep] thresh={th}: {arm(th):.2f} ms/sweep", flush=True)
