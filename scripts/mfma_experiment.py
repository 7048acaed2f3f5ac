#!/usr/bin/env python3
"""settle the MFMA mandate with data.

The most matrix-shaped hot loop is the dense categorical-agreement block of
the PCG-II link weights (GibbsUpdates.scala:370-393): score[r, e] =
sum_a bonus_a(x_{r,a}) * [x == y] over the constant attributes. Runs the
same R x E scorer as (a) a wave-per-record LDS scalar kernel and (b) a bf16
MFMA one-hot matmul (mfma_f32_16x16x32_bf16) with fragments synthesized from
compact codes, verifies they agree, and reports times + effective rates.

Usage (on a GPU box):  python scripts/mfma_experiment.py [R] [E] [iters]
"""
import sys
import time

import numpy as np
import torch

sys.path.insert(0, __file__.rsplit("/", 2)[0])
from dblink_amd import ops

C = ops.native()
DEV = torch.device("cuda", 0)

R = int(sys.argv[1]) if len(sys.argv) > 1 else 16384
E = int(sys.argv[2]) if len(sys.argv) > 2 else 16384
ITERS = int(sys.argv[3]) if len(sys.argv) > 3 else 20
VS = [100, 12, 28]                       # by, bm, bd domain sizes
BASE = np.cumsum([0] + VS)[:3]           # concatenated-domain offsets
K = ((sum(VS) + 31) // 32) * 32          # one-hot width padded for MFMA

rng = np.random.default_rng(0)
rcode = np.stack([rng.integers(0, v, R) + b for v, b in zip(VS, BASE)], 1)
ecode = np.stack([rng.integers(0, v, E) + b for v, b in zip(VS, BASE)], 1)
rbonus = rng.uniform(1.0, 12.0, (R, 3))

t_rc = torch.as_tensor(rcode, dtype=torch.int32, device=DEV).contiguous()
t_ec = torch.as_tensor(ecode, dtype=torch.int32, device=DEV).contiguous()
t_rb = torch.as_tensor(rbonus, dtype=torch.float32, device=DEV).contiguous()
s1 = torch.empty((R, E), dtype=torch.float32, device=DEV)
s2 = torch.empty((R, E), dtype=torch.float32, device=DEV)

C.scalar_score_bench(t_rc, t_rb, t_ec, s1)
C.mfma_score_bench(t_rc, t_rb, t_ec, K, s2)
torch.cuda.synchronize()
# parity (bf16 truncation in the MFMA path: ~2^-8 relative)
diff = (s1 - s2).abs().max().item()
rel = diff / max(s1.abs().max().item(), 1e-9)
print(f"parity: max |scalar - mfma| = {diff:.4f} (rel {rel:.5f})")
assert rel < 2 ** -7, "MFMA path disagrees beyond bf16 truncation"

def bench(fn, label, flops):
    fn()  # warm
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(ITERS):
        fn()
    torch.cuda.synchronize()
    dt = (time.time() - t0) / ITERS
    print(f"{label:>8s}: {dt * 1e3:8.3f} ms  ({flops / dt / 1e12:8.1f} T(FL)OP/s effective)")
    return dt

pairs = R * E
t_sc = bench(lambda: C.scalar_score_bench(t_rc, t_rb, t_ec, s1), "scalar",
             pairs * 9)                   # ~9 VALU ops per pair
t_mf = bench(lambda: C.mfma_score_bench(t_rc, t_rb, t_ec, K, s2), "mfma",
             pairs * K * 2)               # one-hot matmul FLOPs
print(f"pairs/s: scalar {pairs / t_sc / 1e9:.2f} G, mfma {pairs / t_mf / 1e9:.2f} G "
      f"(ratio {t_sc / t_mf:.2f}x in favor of {'mfma' if t_mf < t_sc else 'scalar'})")
