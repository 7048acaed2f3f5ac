"""Linked-pair posterior band: CPU vs GPU (and the all-wave debug value
path) on one dataset — the quick stationary-behavior cross-check."""

import sys, numpy as np
sys.path.insert(0, "/root/repo")
import torch
import bench as b
from dblink_amd.engine.cpu_engine import CpuEngine, SamplerFlags
from dblink_amd.engine.init import deterministic_init
from dblink_amd.parallel.partitioning import KDTreePartitioner

n = 200
cache, rv, rf = b.build_cache_and_records(n, seed=9)
flags = SamplerFlags.for_sampler("PCG-I")
for kind in ("cpu", "gpu", "gpu-allwave"):
    means = []
    for chain_seed in (1, 2, 3):
        p = KDTreePartitioner(0, [])
        st = deterministic_init(rv, rf, np.arange(n, dtype=np.int64), cache, p, seed=chain_seed)
        if kind.startswith("gpu"):
            from dblink_amd.engine.gpu_engine import GpuEngine
            eng = GpuEngine(cache, p, device=torch.device("cuda", 0))
            if kind == "gpu-allwave":
                eng.value_allwave = True
        else:
            eng = CpuEngine(cache, p)
        eng.initial_summary(st)
        pairs = []
        for i in range(300):
            eng.step(st, flags)
            if i >= 100:
                # linked pairs from isolates-free entity counts
                if kind.startswith("gpu"):
                    eng.sync_state(st)
                c = np.bincount(st.rec_ent, minlength=st.num_entities)
                pairs.append(int(np.sum(c * (c - 1) // 2)))
        means.append(np.mean(pairs))
    print(f"{kind}: mean linked pairs per chain = {[round(m,2) for m in means]}, overall {np.mean(means):.2f}", flush=True)
