"""CPU-vs-GPU end-to-end F1 comparison on one synthetic dataset
(validation utility used during development; run on a GPU box)."""

import sys, numpy as np
sys.path.insert(0, "/root/repo")
import torch
import bench as b
from dblink_amd.engine.cpu_engine import CpuEngine, SamplerFlags
from dblink_amd.engine.init import deterministic_init
from dblink_amd.parallel.partitioning import KDTreePartitioner
from dblink_amd.analysis import metrics as mm
from dblink_amd.utils.synthdata import generate

def run_chain(engine_kind, cache, rec_values, rec_files, n, seed):
    p = KDTreePartitioner(1, [3])
    st = deterministic_init(rec_values, rec_files, np.arange(n, dtype=np.int64), cache, p, seed=seed)
    if engine_kind.startswith("gpu"):
        from dblink_amd.engine.gpu_engine import GpuEngine
        eng = GpuEngine(cache, p, device=torch.device("cuda", 0))
        if engine_kind == "gpu-allwave":
            eng.value_allwave = True
    else:
        eng = CpuEngine(cache, p)
    eng.initial_summary(st)
    flags = SamplerFlags.for_sampler("PCG-I")
    # collect link states every 4 iters after 100 burnin
    from collections import Counter
    freq = Counter(); nsamp = 0
    for i in range(500):
        eng.step(st, flags)
        if i >= 100 and i % 4 == 0:
            if engine_kind.startswith("gpu"):
                eng.sync_state(st)
            order = np.argsort(st.rec_ent, kind="stable")
            se = st.rec_ent[order]
            bounds = np.flatnonzero(np.r_[True, se[1:] != se[:-1]])
            nsamp += 1
            for bi, bst in enumerate(bounds):
                e_end = bounds[bi+1] if bi+1 < len(bounds) else len(se)
                gids = frozenset(st.rec_gid[order[bst:e_end]].tolist())
                freq[gids] += 1
    best = {}
    for cl, c in freq.items():
        f = c / nsamp
        for r in cl:
            if r not in best or f > best[r][1]:
                best[r] = (cl, f)
    from collections import defaultdict
    agg = defaultdict(set)
    for r, (cl, _) in best.items():
        agg[cl].add(r)
    return [set(map(str, v)) for v in agg.values()]

for seed in (42, 43, 44):
    n = 300
    cols, _ = generate(n, dup_fraction=0.1, seed=seed)
    cache, rv, rf = b.build_cache_and_records(n, seed=seed)
    truth = mm.membership_to_clusters({str(i): cols["ent_id"][i] for i in range(n)})
    for kind in (["gpu", "gpu-allwave"] if torch.cuda.is_available() else ["cpu"]):
        cl = run_chain(kind, cache, rv, rf, n, seed=319158)
        # rec ids as str(gid)
        pm = mm.PairwiseMetrics.compute(cl, truth)
        print(f"seed={seed} {kind}: P={pm.precision:.3f} R={pm.recall:.3f} F1={pm.f1score:.3f}", flush=True)
