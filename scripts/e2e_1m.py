"""Full project workflow at 1M records on one GPU: sample (with Parquet
linkage writes), summarize, evaluate — wall-clock timed per stage."""
import os, sys, time
sys.path.insert(0, "/root/repo")
import numpy as np
import torch

import bench as b
from dblink_amd.analysis import chain as chain_q
from dblink_amd.analysis import metrics as mm
from dblink_amd.engine import sampler as sampler_m
from dblink_amd.engine.cpu_engine import SamplerFlags
from dblink_amd.engine.gpu_engine import GpuEngine
from dblink_amd.engine.init import deterministic_init
from dblink_amd.parallel.partitioning import KDTreePartitioner
from dblink_amd.utils.synthdata import generate

n = 1_000_000
t0 = time.time()
cols, _ = generate(n, dup_fraction=0.1, seed=77)
cache, rv, rf = b.build_cache_and_records(n, seed=77)
print(f"[e2e] data+cache {time.time()-t0:.1f}s", flush=True)

t0 = time.time()
part = KDTreePartitioner(6, [3, 4])  # 64 partitions: BASELINE config-3 granularity
state = deterministic_init(rv, rf, np.arange(n, dtype=np.int64), cache, part, seed=319158)
engine = GpuEngine(cache, part, device=torch.device("cuda", 0))
engine.rec_ids_array = cols["rec_id"].astype(object)
engine.initial_summary(state)
print(f"[e2e] init {time.time()-t0:.1f}s", flush=True)

out = "/tmp/e2e_1m"
os.makedirs(out, exist_ok=True)
t0 = time.time()
sampler_m.sample(engine, state, sample_size=50, output_path=out,
                 burnin_interval=400, thinning_interval=4, checkpoint_interval=0,
                 flags=SamplerFlags.for_sampler("PCG-I"))
dt = time.time() - t0
print(f"[e2e] sample 600 iterations + 50 sample writes: {dt:.1f}s "
      f"({600/dt:.1f} it/s incl. writes)", flush=True)

t0 = time.time()
table = chain_q.load_chain(out, 400)
print(f"[e2e] chain load: {time.time()-t0:.1f}s rows={table.num_rows}", flush=True)
t0 = time.time()
smpc = chain_q.shared_most_probable_clusters_fast(table)
print(f"[e2e] fast sMPC: {time.time()-t0:.1f}s clusters={len(smpc)}", flush=True)
t0 = time.time()
truth = mm.membership_to_clusters({cols["rec_id"][i]: cols["ent_id"][i] for i in range(n)})
pm = mm.PairwiseMetrics.compute(smpc, truth)
print(f"[e2e] metrics: {time.time()-t0:.1f}s P={pm.precision:.3f} R={pm.recall:.3f} "
      f"F1={pm.f1score:.3f}", flush=True)
