// CDNA4 (gfx950) HIP kernels for the partitioned Gibbs sweep.
//
// Kernel inventory (SURVEY.md §2.4 K-list):
//   K3+K4+K5  link_update           — posting-list intersection + per-candidate
//                                     log-weights + fused Gumbel-max draw
//             link_update_dense     — PCG-II collapsed / Gibbs-Sequential dense
//                                     variants (LDS-tiled over entity values)
//   K6        value_update          — per-(entity, attribute) collapsed /
//                                     non-collapsed value draw with sparse
//                                     perturbation weights merged in an LDS
//                                     hash table; alias-table base draws
//             value_update_seq      — brute-force dense-domain variant
//   K7        distortion_update     — element-wise Bernoulli resample (Philox)
//   K8        summary_loglik        — f64 block/atomic reduction of the
//                                     log-likelihood terms
//   K9a       kd_descent            — flat KD-tree partition reassignment
//   K1 (GPU)  sim_pairs count/fill  — banded Levenshtein domain sweep
//
// One wave (64 lanes) owns one record / one (entity, attribute) pair; block
// size 256 = 4 waves. All categorical draws use Gumbel-max with Philox
// counter RNG (common.h), so results are independent of scheduling order.

#include <ATen/cuda/CUDAContext.h>
#include <hip/hip_runtime.h>
#include <rocprim/device/device_radix_sort.hpp>
#include <torch/extension.h>

#include "common.h"

namespace dblink {

#define CHECK_GPU(x) TORCH_CHECK(x.is_cuda(), #x " must be a GPU tensor")
#define CHECK_CONTIG(x) TORCH_CHECK(x.is_contiguous(), #x " must be contiguous")

constexpr int MAX_ATTRS = 16;
constexpr uint32_t PH_LINK = 1, PH_DIST = 2, PH_VALG = 3, PH_VALM = 4,
                   PH_LINKH = 7, PH_LINKF = 8;

// ---------------------------------------------------------------------------
// Staged similarity rows. A record's od-attribute CSR sim rows are FIXED for
// the whole candidate scan, yet sim_lookup pays a log2(row)-deep
// dependent-load binary search per (candidate, attr). PMC counters
// (profiles/pmc_r02_1M64.txt) show the link kernels latency-bound
// (SQ_WAIT_ANY >> SQ_ACTIVE_INST), so that chain is the bottleneck: staging
// each row once into a per-wave LDS open-addressing hash turns it into ~1
// LDS probe. Sub-table descriptors are nibble-packed per attr (runtime-
// indexed local arrays would spill to scratch): logpk nibble = log2(slots)
// (0 = not staged -> global binary search, 15 = empty row -> sim 0), offpk
// nibble = offset/32. Load factor <= 1/2; rows that don't fit stay global.
// ---------------------------------------------------------------------------
constexpr int SIMH_CAP = 512;  // staged entries per wave (4 KB key+value)

__device__ inline void simh_stage(
    int32_t* kh, float* vh, int lane, const int64_t* __restrict__ csr_row_ptr,
    const int32_t* __restrict__ csr_col, const float* __restrict__ csr_sim,
    const int64_t* __restrict__ voff, const int32_t* __restrict__ rec_values,
    int64_t r, int A, uint32_t od_mask, uint64_t& logpk, uint64_t& offpk) {
  logpk = 0;
  offpk = 0;
  int cursor = 0;
  for (uint32_t m = od_mask; m;) {
    const int a = __ffs(m) - 1;
    m &= m - 1;
    const int64_t row = voff[a] + rec_values[r * A + a];
    const int64_t lo = csr_row_ptr[row];
    const int len = (int)(csr_row_ptr[row + 1] - lo);
    if (len == 0) {
      logpk |= 15ull << (4 * a);
      continue;
    }
    int lg = 5;  // >= 32 slots keeps offsets on /32 nibble boundaries
    while ((1 << lg) < 2 * len) ++lg;
    if (lg > 9 || cursor + (1 << lg) > SIMH_CAP) continue;  // stays global
    const int sz = 1 << lg, base = cursor;
    for (int i = lane; i < sz; i += WAVE) kh[base + i] = -1;
    // wave-lockstep: the CAS claims below see the -1 fill of this wave
    for (int i = lane; i < len; i += WAVE) {
      const int32_t y = csr_col[lo + i];
      int h = (int)(((uint32_t)y * 2654435761u) >> (32 - lg));
      for (;;) {
        const int32_t prev = atomicCAS(&kh[base + h], -1, y);
        if (prev == -1) {
          vh[base + h] = csr_sim[lo + i];
          break;
        }
        h = (h + 1) & (sz - 1);
      }
    }
    logpk |= (uint64_t)lg << (4 * a);
    offpk |= (uint64_t)(base >> 5) << (4 * a);
    cursor += sz;
  }
}

__device__ inline float simh_lookup(const int32_t* kh, const float* vh,
                                    uint64_t logpk, uint64_t offpk, int a,
                                    const int64_t* __restrict__ csr_row_ptr,
                                    const int32_t* __restrict__ csr_col,
                                    const float* __restrict__ csr_sim,
                                    int64_t row, int32_t y) {
  const int lg = (int)((logpk >> (4 * a)) & 15ull);
  if (lg == 15) return 0.0f;  // empty sim row
  if (lg == 0)                // row not staged
    return sim_lookup(csr_row_ptr, csr_col, csr_sim, row, y);
  const int base = (int)(((offpk >> (4 * a)) & 15ull) << 5);
  int h = (int)(((uint32_t)y * 2654435761u) >> (32 - lg));
  const int msk = (1 << lg) - 1;
  for (;;) {
    const int32_t k = kh[base + h];
    if (k == y) return vh[base + h];
    if (k == -1) return 0.0f;  // staged table holds the FULL row
    h = (h + 1) & msk;
  }
}

// ---------------------------------------------------------------------------
// K3+K4+K5: link update (PCG-I / Gibbs indexed path)
//
// Posting lists are used ONLY to enumerate the smallest candidate set;
// membership in every other non-distorted attribute's posting list is
// equivalent to a direct entity-value comparison (the posting list of
// (partition, attr, v) is exactly the entities whose attr equals v), so no
// binary search / bitmap is needed and posting order is irrelevant — which
// is what lets the index build be an unstable counting sort.
// ---------------------------------------------------------------------------

__global__ void link_update_kernel(
    const int32_t* __restrict__ rec_values,  // [R, A]
    const uint8_t* __restrict__ rec_dist,    // [R, A]
    const int64_t* __restrict__ rec_gid,     // [R]
    const int32_t* __restrict__ rec_part,    // [R]
    const int64_t* __restrict__ cand_lo,     // [R, A]
    const int64_t* __restrict__ cand_hi,     // [R, A]
    const int32_t* __restrict__ postings,    // [E*A]
    const int32_t* __restrict__ ent_values,  // [E, A]
    const int64_t* __restrict__ ent_ptr,     // [P+1]
    const float* __restrict__ log_norm,      // [Vtot]
    const int64_t* __restrict__ voff,        // [A+1]
    const int64_t* __restrict__ csr_row_ptr, // [Vtot+1]
    const int32_t* __restrict__ csr_col,
    const float* __restrict__ csr_sim,
    const uint8_t* __restrict__ attr_const,  // [A]
    const int32_t* __restrict__ pair_a1,     // [NP] const-pair pseudo slots
    const int32_t* __restrict__ pair_a2, int NP,
    const uint8_t* __restrict__ small_mask,  // [R] 1 = handled by small kernel (or null)
    int64_t R, int A,
    uint64_t seed, uint32_t iteration, const int64_t* __restrict__ ctrl,
    int64_t* __restrict__ rec_ent_out,       // [R]
    const int64_t* __restrict__ rec_ent_in,  // [R]
    int* __restrict__ error_count, int simh_on) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  __shared__ int32_t simh_k_s[4][SIMH_CAP];  // launcher uses WPB == 4
  __shared__ float simh_v_s[4][SIMH_CAP];
  int64_t r = (int64_t)blockIdx.x * (blockDim.x / WAVE) + wid;
  if (r >= R) return;
  if (small_mask != nullptr && small_mask[r]) return;
  if (ctrl != nullptr) { seed = (uint64_t)ctrl[0]; iteration = (uint32_t)ctrl[1]; }

  // Attribute classification as BITMASKS, not arrays: runtime-indexed local
  // arrays spill to scratch (measured 208 B/lane = a scratch round-trip per
  // membership probe). All rec_values/voff reads below are wave-uniform, so
  // they compile to scalar-cache loads and cost nothing per candidate.
  // Constant observed-distorted attrs scale all weights equally and cancel
  // under normalization, so od_mask keeps only non-constant ones.
  const int T = A + NP;
  uint32_t nd_mask = 0, od_mask = 0;
  for (int a = 0; a < A; ++a) {
    const int32_t x = rec_values[r * A + a];
    if (x < 0) continue;
    if (!rec_dist[r * A + a]) nd_mask |= 1u << a;
    else if (!attr_const[a]) od_mask |= 1u << a;
  }
  // base = smallest candidate list among nd attrs and (optional) const-pair
  // pseudo slots; a pair base implies both constituents match, but every nd
  // single must still be value-checked (check_mask keeps them all).
  int base_attr = -1;
  int64_t base_lo = 0, base_n = INT64_MAX;
  for (uint32_t m = nd_mask; m;) {
    const int a = __ffs(m) - 1;
    m &= m - 1;
    const int64_t lo = cand_lo[r * T + a], n = cand_hi[r * T + a] - lo;
    if (n < base_n) { base_n = n; base_lo = lo; base_attr = a; }
  }
  for (int t = 0; t < NP; ++t) {
    const int a1 = pair_a1[t], a2 = pair_a2[t];
    if (((nd_mask >> a1) & 1u) == 0 || ((nd_mask >> a2) & 1u) == 0) continue;
    const int64_t lo = cand_lo[r * T + A + t], n = cand_hi[r * T + A + t] - lo;
    if (n < base_n) { base_n = n; base_lo = lo; base_attr = -1; }
  }
  bool base_postings = true;
  if (nd_mask == 0) {
    const int32_t p = rec_part[r];
    base_lo = ent_ptr[p];
    base_n = ent_ptr[p + 1] - base_lo;
    base_postings = false;
  }
  const uint32_t check_mask =
      base_attr >= 0 ? (nd_mask & ~(1u << base_attr)) : nd_mask;

  const uint64_t gid = (uint64_t)rec_gid[r];
  // stage this record's od-attr sim rows into LDS when the scan is long
  // enough to amortize the fill (short scans: the binary search is cheaper)
  uint64_t logpk = 0, offpk = 0;
  int32_t* simh_k = simh_k_s[wid];
  float* simh_v = simh_v_s[wid];
  if (simh_on && od_mask && base_n * __popc(od_mask) >= 32)
    simh_stage(simh_k, simh_v, lane, csr_row_ptr, csr_col, csr_sim, voff,
               rec_values, r, A, od_mask, logpk, offpk);
  float best_score = -INFINITY;
  long long best_e = -1;
  for (int64_t i = lane; i < base_n; i += WAVE) {
    int32_t e = base_postings ? postings[base_lo + i] : (int32_t)(base_lo + i);
    bool ok = true;
    for (uint32_t m = check_mask; m;) {
      const int a = __ffs(m) - 1;
      m &= m - 1;
      if (ent_values[(int64_t)e * A + a] != rec_values[r * A + a]) {
        ok = false;
        break;
      }
    }
    if (!ok) continue;
    float logw = 0.0f;
    for (uint32_t m = od_mask; m;) {
      const int a = __ffs(m) - 1;
      m &= m - 1;
      const int32_t y = ent_values[(int64_t)e * A + a];
      logw += log_norm[voff[a] + y] +
              simh_lookup(simh_k, simh_v, logpk, offpk, a, csr_row_ptr,
                          csr_col, csr_sim, voff[a] + rec_values[r * A + a], y);
    }
    float g = gumbel_from_uniform(
        philox_uniform(seed, iteration, PH_LINK, gid, (uint32_t)e));
    float score = logw + g;
    if (score > best_score) { best_score = score; best_e = e; }
  }
  wave_argmax(best_score, best_e);
  if (lane == 0) {
    if (best_e < 0) {  // empty candidate set: state invariant violated
      atomicAdd(error_count, 1);
      best_e = rec_ent_in[r];
    }
    rec_ent_out[r] = best_e;
  }
}

// Thread-per-record link update for records whose smallest candidate list is
// short (the common case once clusters localize): one thread walks the
// intersection serially — 64x fewer wave slots than the wave path.
__global__ void link_update_small_kernel(
    const uint8_t* __restrict__ small_mask, int64_t n_recs,
    const int32_t* __restrict__ rec_values, const uint8_t* __restrict__ rec_dist,
    const int64_t* __restrict__ rec_gid, const int64_t* __restrict__ cand_lo,
    const int64_t* __restrict__ cand_hi, const int32_t* __restrict__ postings,
    const int32_t* __restrict__ ent_values, const float* __restrict__ log_norm,
    const int64_t* __restrict__ voff, const int64_t* __restrict__ csr_row_ptr,
    const int32_t* __restrict__ csr_col, const float* __restrict__ csr_sim,
    const uint8_t* __restrict__ attr_const,
    const int32_t* __restrict__ pair_a1,
    const int32_t* __restrict__ pair_a2, int NP, int A, uint64_t seed,
    uint32_t iteration, const int64_t* __restrict__ ctrl,
    int64_t* __restrict__ rec_ent_out,
    const int64_t* __restrict__ rec_ent_in, int* __restrict__ error_count) {
  const int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= n_recs) return;
  if (small_mask[idx] != 1) return;
  if (ctrl != nullptr) { seed = (uint64_t)ctrl[0]; iteration = (uint32_t)ctrl[1]; }
  const int64_t r = idx;

  // Bitmask classification (see link_update_kernel: local arrays spill)
  const int T = A + NP;
  uint32_t nd_mask = 0, od_mask = 0;
  for (int a = 0; a < A; ++a) {
    const int32_t x = rec_values[r * A + a];
    if (x < 0) continue;
    if (!rec_dist[r * A + a]) nd_mask |= 1u << a;
    else if (!attr_const[a]) od_mask |= 1u << a;
  }
  int base_attr = -1;
  int64_t base_lo = 0, base_sz = INT64_MAX;
  for (uint32_t m = nd_mask; m;) {
    const int a = __ffs(m) - 1;
    m &= m - 1;
    const int64_t lo = cand_lo[r * T + a], n = cand_hi[r * T + a] - lo;
    if (n < base_sz) { base_sz = n; base_lo = lo; base_attr = a; }
  }
  for (int t = 0; t < NP; ++t) {
    const int a1 = pair_a1[t], a2 = pair_a2[t];
    if (((nd_mask >> a1) & 1u) == 0 || ((nd_mask >> a2) & 1u) == 0) continue;
    const int64_t lo = cand_lo[r * T + A + t], n = cand_hi[r * T + A + t] - lo;
    if (n < base_sz) { base_sz = n; base_lo = lo; base_attr = -1; }
  }
  const uint32_t check_mask =
      base_attr >= 0 ? (nd_mask & ~(1u << base_attr)) : nd_mask;
  const uint64_t gid = (uint64_t)rec_gid[r];
  float best_score = -INFINITY;
  long long best_e = -1;
  for (int64_t i = base_lo; i < base_lo + base_sz; ++i) {
    const int32_t e = postings[i];
    bool ok = true;
    for (uint32_t m = check_mask; m;) {
      const int a = __ffs(m) - 1;
      m &= m - 1;
      if (ent_values[(int64_t)e * A + a] != rec_values[r * A + a]) {
        ok = false;
        break;
      }
    }
    if (!ok) continue;
    float logw = 0.0f;
    for (uint32_t m = od_mask; m;) {
      const int a = __ffs(m) - 1;
      m &= m - 1;
      const int32_t y = ent_values[(int64_t)e * A + a];
      logw += log_norm[voff[a] + y] +
              sim_lookup(csr_row_ptr, csr_col, csr_sim,
                         voff[a] + rec_values[r * A + a], y);
    }
    const float g = gumbel_from_uniform(
        philox_uniform(seed, iteration, PH_LINK, gid, (uint32_t)e));
    if (logw + g > best_score) { best_score = logw + g; best_e = e; }
  }
  if (best_e < 0) {
    atomicAdd(error_count, 1);
    best_e = rec_ent_in[r];
  }
  rec_ent_out[r] = best_e;
}

// ---------------------------------------------------------------------------
// Heavy link update: hierarchical (A*) Gumbel-max sampler.
//
// In the model's stationary regime the distortion probabilities are large
// (~0.2-0.6 under the shipped demo priors, measured on real RLdata10000), so
// many records have few observed NON-distorted attributes: their candidate
// set is a whole low-cardinality posting range (thousands of entities) or
// the entire partition (GibbsUpdates.scala:398-430 with an empty
// obsNonDistorted set). Scanning those ranges is what made sweeps grow
// ~100x between burn-in and stationarity. This kernel draws the SAME
// categorical (exactly — validated statistically against the scan path)
// in O(|similar set| + tens) per record:
//
//   score(e) = t(e) + sim(e) + gumbel_e,   t(e) = sum_od lognorm(y_e,a) <= 0
//
//   1. exact Gumbel-max over S_r = entities similar to any observed-distorted
//      record value (enumerated via the sim-row's posting segments) that pass
//      the non-distorted equality checks; per-entity keyed gumbels.
//   2. the complement (sim == 0) via A* sampling with a UNIFORM proposal
//      (q = 0 >= t since every log-normalizer is <= 0): successive maxima
//      G_1 > G_2 > ... of the pool's gumbels via truncated-Gumbel draws;
//      each drawn entity is excluded (Z -= 1) and, if it passes the checks
//      and is not similar, scores t(e) + G_k. Stop when G <= current best
//      (no remaining candidate can win: t <= 0).
//   3. winner = argmax of the two halves. Independent keyed RNG streams per
//      half keep the combined draw an exact categorical sample.
//
// All draws are keyed by (seed, iteration, record gid, counter), so results
// are deterministic and independent of posting order. If the A* loop hits
// its exclusion cap (rare; counted in stats[1]), the record falls back to a
// deterministic full scan with a FRESH key stream (PH_LINKF) — exact
// regardless of what the truncated chain did.
// ---------------------------------------------------------------------------

constexpr int HEAVY_EXC_CAP = 512;   // A* exclusions per record before fallback
constexpr int HEAVY_HASH = 1024;     // LDS exclusion hash slots per wave
constexpr int HEAVY_WAVES = 4;

__global__ void link_update_heavy_kernel(
    const uint8_t* __restrict__ mode,        // [R] 2 = heavy
    const int32_t* __restrict__ rec_values, const uint8_t* __restrict__ rec_dist,
    const int64_t* __restrict__ rec_gid, const int32_t* __restrict__ rec_part,
    const int32_t* __restrict__ ent_values, const int64_t* __restrict__ ent_ptr,
    const float* __restrict__ log_norm, const int64_t* __restrict__ voff,
    const int64_t* __restrict__ csr_row_ptr, const int32_t* __restrict__ csr_col,
    const float* __restrict__ csr_sim, const uint8_t* __restrict__ attr_const,
    const int64_t* __restrict__ csr_row_ptr_big,
    const int32_t* __restrict__ csr_col_big, const float* __restrict__ csr_sim_big,
    float tau,
    const int32_t* __restrict__ postings, const int64_t* __restrict__ idx_ptr,
    int64_t Vmax, int NP, int64_t R, int A,
    uint64_t seed, uint32_t iteration, const int64_t* __restrict__ ctrl,
    int64_t* __restrict__ rec_ent_out, const int64_t* __restrict__ rec_ent_in,
    int* __restrict__ error_count,
    unsigned long long* __restrict__ stats, int simh_on) {
  // stats[4] (or null): {A* iterations, fallbacks, S_r visits, heavy records}
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  __shared__ int32_t simh_k_s[HEAVY_WAVES][SIMH_CAP];
  __shared__ float simh_v_s[HEAVY_WAVES][SIMH_CAP];
  const int64_t r = (int64_t)blockIdx.x * HEAVY_WAVES + wid;
  if (r >= R || mode[r] != 2) return;
  if (ctrl != nullptr) { seed = (uint64_t)ctrl[0]; iteration = (uint32_t)ctrl[1]; }

  const int T = A + NP;
  uint32_t nd_mask = 0, od_mask = 0;
  for (int a = 0; a < A; ++a) {
    const int32_t x = rec_values[r * A + a];
    if (x < 0) continue;
    if (!rec_dist[r * A + a]) nd_mask |= 1u << a;
    else if (!attr_const[a]) od_mask |= 1u << a;
  }
  const int32_t p = rec_part[r];
  const int64_t e0 = ent_ptr[p], e1 = ent_ptr[p + 1];
  const uint64_t gid = (uint64_t)rec_gid[r];

  // proposal pool: the whole partition (no nd attr) or the base posting
  // segment of the single nd attr (postings are stable-sorted when this
  // kernel is active, so indexed draws are deterministic)
  const bool seg_pool = nd_mask != 0;
  int64_t plo, pn;
  if (seg_pool) {
    const int a = __ffs(nd_mask) - 1;
    const int64_t key = ((int64_t)p * T + a) * Vmax + rec_values[r * A + a];
    plo = idx_ptr[key];
    pn = idx_ptr[key + 1] - plo;
  } else {
    plo = e0;
    pn = e1 - e0;
  }

  // heavy scans are always long: stage the record's od-attr sim rows in LDS
  uint64_t logpk = 0, offpk = 0;
  int32_t* simh_k = simh_k_s[wid];
  float* simh_v = simh_v_s[wid];
  if (simh_on && od_mask)
    simh_stage(simh_k, simh_v, lane, csr_row_ptr, csr_col, csr_sim, voff,
               rec_values, r, A, od_mask, logpk, offpk);

  // ---- 1. exact scan over the similar set S_r ----------------------------
  // The sim-row's posting segments average only a few entries, so a
  // segment-at-a-time wave loop would idle ~60 of 64 lanes. Instead each
  // 64-column chunk is flattened warp-cooperatively: a register prefix scan
  // of the segment lengths, then lanes stride the flat space and locate
  // their owning column by a shuffle-based binary search.
  float best_f = -INFINITY;
  long long best_e = -1;
  for (uint32_t m = od_mask; m;) {
    const int a = __ffs(m) - 1;
    m &= m - 1;
    const int64_t row = voff[a] + rec_values[r * A + a];
    const int64_t j0 = csr_row_ptr_big[row], j1 = csr_row_ptr_big[row + 1];
    for (int64_t cbase = j0; cbase < j1; cbase += WAVE) {
      const int64_t myj = cbase + lane;
      long long lo = 0;
      int n = 0;
      float simv = 0.0f;
      if (myj < j1) {
        const int64_t key = ((int64_t)p * T + a) * Vmax + csr_col_big[myj];
        lo = idx_ptr[key];
        n = (int)(idx_ptr[key + 1] - lo);
        simv = csr_sim_big[myj];
      }
      int pre = n;  // inclusive prefix of segment lengths across lanes
#pragma unroll
      for (int off = 1; off < WAVE; off <<= 1) {
        const int v = __shfl_up(pre, off);
        if (lane >= off) pre += v;
      }
      const int total = __shfl(pre, WAVE - 1);
      const int excl = pre - n;
      for (int sb = 0; sb < total; sb += WAVE) {
        const int s = sb + lane;
        // owning lane = last lane with excl <= s (shuffle binary search);
        // every lane participates in the shuffles (uniform loop bound)
        int fl = 0;
#pragma unroll
        for (int step = WAVE / 2; step; step >>= 1) {
          const int cand = fl + step;
          const int ce = __shfl(excl, cand < WAVE ? cand : WAVE - 1);
          if (cand < WAVE && ce <= s) fl = cand;
        }
        const int64_t i = __shfl(lo, fl) + (s - __shfl(excl, fl));
        const float dsim = __shfl(simv, fl);  // sim of the enumerated attr
        if (s >= total) continue;
        const int32_t e = postings[i];
        bool ok = true;
        for (uint32_t mm = nd_mask; mm;) {
          const int aa = __ffs(mm) - 1;
          mm &= mm - 1;
          if (ent_values[(int64_t)e * A + aa] != rec_values[r * A + aa]) {
            ok = false;
            break;
          }
        }
        if (!ok) continue;
        float logw = 0.0f;
        for (uint32_t mo = od_mask; mo;) {
          const int ao = __ffs(mo) - 1;
          mo &= mo - 1;
          const int32_t y = ent_values[(int64_t)e * A + ao];
          logw += log_norm[voff[ao] + y] +
                  (ao == a ? dsim
                           : simh_lookup(simh_k, simh_v, logpk, offpk, ao,
                                         csr_row_ptr, csr_col, csr_sim,
                                         voff[ao] + rec_values[r * A + ao], y));
        }
        const float g = gumbel_from_uniform(
            philox_uniform(seed, iteration, PH_LINK, gid, (uint32_t)e));
        if (logw + g > best_f) { best_f = logw + g; best_e = e; }
      }
      if (lane == 0 && stats != nullptr)
        atomicAdd(&stats[2], (unsigned long long)total);
    }
  }
  wave_argmax(best_f, best_e);

  // ---- 2. A* over the complement (lane 0 serial) -------------------------
  // Proposal: uniform with constant shift q = |od| * tau — every pool entity
  // (all big-sims excluded) has score t + sum(small sims) + g <= q + g, so
  // the successive proposal maxima q + G_k bound the remaining candidates.
  __shared__ int32_t exclh_s[HEAVY_WAVES][HEAVY_HASH];
  __shared__ long long winner_s[HEAVY_WAVES];
  __shared__ int fell_back_s[HEAVY_WAVES];
  int32_t* exclh = exclh_s[wid];
  for (int i = lane; i < HEAVY_HASH; i += WAVE) exclh[i] = -1;
  // wave-lockstep: lane 0 reads below see these LDS writes
  if (lane == 0) {
    const double q_shift = (double)tau * (double)__popc(od_mask);
    double best = (best_e >= 0) ? (double)best_f : -INFINITY;
    long long bE = best_e;
    double Z = (double)pn;
    double b = 0.0;
    bool first = true;
    uint32_t ctr = 0;
    int n_exc = 0;
    bool fell_back = false;
    unsigned long long iters = 0;
    while (Z > 0.5) {
      const double logZ = log(Z);
      double G;
      {
        const double u = (double)philox_uniform(seed, iteration, PH_LINKH, gid, ctr++);
        G = first ? logZ - log(-log(u)) : logZ - log(exp(logZ - b) - log(u));
        first = false;
      }
      b = G;
      if (G + q_shift <= best) break;
      // categorical (uniform) draw over pool \ excluded, by pool index;
      // the single-writer LDS hash both tests and records exclusions
      int64_t ci = -1;
      for (int tries = 0; tries < 4 * HEAVY_EXC_CAP; ++tries) {
        const float u = philox_uniform(seed, iteration, PH_LINKH, gid, ctr++);
        int64_t cand = (int64_t)(u * (float)pn);
        if (cand >= pn) cand = pn - 1;
        int h = (int)(((uint32_t)cand * 2654435761u) >> 16) & (HEAVY_HASH - 1);
        bool fresh = false;
        for (;;) {
          const int32_t cur = exclh[h];
          if (cur == (int32_t)cand) break;          // already excluded
          if (cur == -1) { exclh[h] = (int32_t)cand; fresh = true; break; }
          h = (h + 1) & (HEAVY_HASH - 1);
        }
        if (fresh) { ci = cand; break; }
      }
      if (ci < 0 || ++n_exc > HEAVY_EXC_CAP) { fell_back = true; break; }
      Z -= 1.0;
      ++iters;
      const int32_t e = seg_pool ? postings[plo + ci] : (int32_t)(plo + ci);
      bool ok = true;
      for (uint32_t mm = nd_mask; mm;) {
        const int aa = __ffs(mm) - 1;
        mm &= mm - 1;
        if (ent_values[(int64_t)e * A + aa] != rec_values[r * A + aa]) {
          ok = false;
          break;
        }
      }
      if (!ok) continue;
      double t = 0.0;
      bool in_big = false;
      for (uint32_t mo = od_mask; mo;) {
        const int ao = __ffs(mo) - 1;
        mo &= mo - 1;
        const int32_t y = ent_values[(int64_t)e * A + ao];
        const float s = simh_lookup(simh_k, simh_v, logpk, offpk, ao,
                                    csr_row_ptr, csr_col, csr_sim,
                                    voff[ao] + rec_values[r * A + ao], y);
        if (s >= tau) { in_big = true; break; }  // scored exactly in S_r
        t += (double)(log_norm[voff[ao] + y] + s);
      }
      if (in_big) continue;
      const double s_true = t + G;  // gumbel realization of this draw == G
      if (s_true > best) { best = s_true; bE = e; }
    }
    winner_s[wid] = bE;
    fell_back_s[wid] = fell_back ? 1 : 0;
    if (stats != nullptr) {
      atomicAdd(&stats[0], iters);
      if (fell_back) atomicAdd(&stats[1], 1ull);
      atomicAdd(&stats[3], 1ull);  // heavy records processed
    }
  }
  __syncthreads();

  // ---- 3. rare fallback: deterministic full scan, fresh key stream -------
  if (fell_back_s[wid]) {
    float bf = -INFINITY;
    long long be = -1;
    for (int64_t e = e0 + lane; e < e1; e += WAVE) {
      bool ok = true;
      for (uint32_t mm = nd_mask; mm;) {
        const int aa = __ffs(mm) - 1;
        mm &= mm - 1;
        if (ent_values[e * A + aa] != rec_values[r * A + aa]) { ok = false; break; }
      }
      if (!ok) continue;
      float logw = 0.0f;
      for (uint32_t mo = od_mask; mo;) {
        const int ao = __ffs(mo) - 1;
        mo &= mo - 1;
        const int32_t y = ent_values[e * A + ao];
        logw += log_norm[voff[ao] + y] +
                simh_lookup(simh_k, simh_v, logpk, offpk, ao, csr_row_ptr,
                            csr_col, csr_sim,
                            voff[ao] + rec_values[r * A + ao], y);
      }
      const float g = gumbel_from_uniform(
          philox_uniform(seed, iteration, PH_LINKF, gid, (uint32_t)(e - e0)));
      if (logw + g > bf) { bf = logw + g; be = e; }
    }
    wave_argmax(bf, be);
    if (lane == 0) winner_s[wid] = be;
  }
  if (lane == 0) {
    long long w = winner_s[wid];
    if (w < 0) {  // empty candidate set: state invariant violated
      atomicAdd(error_count, 1);
      w = rec_ent_in[r];
    }
    rec_ent_out[r] = w;
  }
}

// ---------------------------------------------------------------------------
// Dense link update: PCG-II (collapsed) and Gibbs-Sequential
// ---------------------------------------------------------------------------

__global__ void link_update_dense_kernel(
    const int32_t* __restrict__ rec_values, const uint8_t* __restrict__ rec_dist,
    const int64_t* __restrict__ rec_gid, const int32_t* __restrict__ rec_part,
    const int32_t* __restrict__ rec_file,
    const int32_t* __restrict__ ent_values, const int64_t* __restrict__ ent_ptr,
    const float* __restrict__ theta,       // [A, F]
    const float* __restrict__ phi,         // [Vtot] linear probability
    const float* __restrict__ norm_lin,    // [Vtot] linear 1/normalizer
    const int64_t* __restrict__ voff, const int64_t* __restrict__ csr_row_ptr,
    const int32_t* __restrict__ csr_col, const float* __restrict__ csr_sim,
    const uint8_t* __restrict__ attr_const, int64_t R, int A, int F,
    int collapsed,  // 1 = PCG-II weights, 0 = Gibbs-Sequential weights
    uint64_t seed, uint32_t iteration, const int64_t* __restrict__ ctrl,
    int64_t* __restrict__ rec_ent_out, int simh_on) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  __shared__ int32_t simh_k_s[4][SIMH_CAP];  // launcher uses WPB == 4
  __shared__ float simh_v_s[4][SIMH_CAP];
  const int64_t r = (int64_t)blockIdx.x * (blockDim.x / WAVE) + wid;
  if (r >= R) return;
  if (ctrl != nullptr) { seed = (uint64_t)ctrl[0]; iteration = (uint32_t)ctrl[1]; }
  const int32_t p = rec_part[r];
  const int64_t e0 = ent_ptr[p], e1 = ent_ptr[p + 1];
  const int f = rec_file[r];
  const uint64_t gid = (uint64_t)rec_gid[r];

  // stage the record's sim rows once per wave: the dense scan pays a
  // sim_lookup per (entity, attr), all against rows fixed for this record
  uint32_t st_mask = 0;
  for (int a = 0; a < A; ++a) {
    if (rec_values[r * A + a] < 0 || attr_const[a]) continue;
    if (collapsed || rec_dist[r * A + a]) st_mask |= 1u << a;
  }
  uint64_t logpk = 0, offpk = 0;
  int32_t* simh_k = simh_k_s[wid];
  float* simh_v = simh_v_s[wid];
  if (simh_on && st_mask && (e1 - e0) * __popc(st_mask) >= 32)
    simh_stage(simh_k, simh_v, lane, csr_row_ptr, csr_col, csr_sim, voff,
               rec_values, r, A, st_mask, logpk, offpk);

  float best_score = -INFINITY;
  long long best_e = -1;
  for (int64_t e = e0 + lane; e < e1; e += WAVE) {
    float logw = 0.0f;
    for (int a = 0; a < A; ++a) {
      const int32_t x = rec_values[r * A + a];
      if (x < 0) continue;
      const int32_t y = ent_values[e * A + a];
      if (collapsed) {
        const float th = theta[a * F + f];
        float like = phi[voff[a] + x];
        if (!attr_const[a]) {
          float s = simh_lookup(simh_k, simh_v, logpk, offpk, a, csr_row_ptr,
                                csr_col, csr_sim, voff[a] + x, y);
          like *= norm_lin[voff[a] + y] * __expf(s);
        }
        float w = (y == x ? 1.0f - th : 0.0f) + th * like;
        logw += __logf(w);
      } else {
        // Gibbs-Sequential: non-distorted must match exactly
        if (!rec_dist[r * A + a]) {
          if (x != y) { logw = -INFINITY; break; }
        } else if (!attr_const[a]) {
          float s = simh_lookup(simh_k, simh_v, logpk, offpk, a, csr_row_ptr,
                                csr_col, csr_sim, voff[a] + x, y);
          logw += __logf(norm_lin[voff[a] + y]) + s;  // phi(x) constant: cancels
        }
      }
    }
    if (logw == -INFINITY) continue;
    float g = gumbel_from_uniform(
        philox_uniform(seed, iteration, PH_LINK, gid, (uint32_t)(e - e0)));
    float score = logw + g;
    if (score > best_score) { best_score = score; best_e = e; }
  }
  wave_argmax(best_score, best_e);
  if (lane == 0) rec_ent_out[r] = best_e;  // weights always > 0 when collapsed
}

// ---------------------------------------------------------------------------
// K6: value update
// ---------------------------------------------------------------------------

// LDS hash table per wave for sparse perturbation weights.
constexpr int HASH_CAP = 1024;            // slots per wave
constexpr int WAVES_PER_BLOCK_VAL = 4;    // 256 threads
constexpr int VAL_DMAX = 16;              // distinct (value, file) groups per pair

struct ValueArgs {
  const int64_t* ctrl;         // [2] = {seed, iteration} device override (or null)
  const int64_t* pair_list;    // [n_pairs] flattened (e*A + a) (or null)
  const int32_t* kobs;         // [E*A] observed-linked counts for self-selection (or null)
  int64_t n_pairs;
  const double* csr_excl;      // [nnz] exclusive row prefix of raw k=1 weights
  const double* csr_rawsum;    // [Vtot] row totals of raw k=1 weights
  const double* z1;            // [A] power-1 normalizer Z_1
  // k >= 2 single-distinct-value tables (set_value_ktables): level k-2
  const double* tab_excl;      // [(kmax-1) * nnz] exclusive row prefixes
  const double* tab_rawsum;    // [(kmax-1) * Vtot] row totals
  const float* self_expsim;    // [Vtot]
  int ktab_max;                // largest k covered (0 = disabled)
  int64_t nnz;
  // 2-distinct-value (k, m) tables: level (k-2)(k-1)/2 + m - 1
  const double* tab2_excl;
  const double* tab2_rawsum;
  int k2tab_max;
  const int32_t* rec_values;
  const uint8_t* rec_dist;
  const int32_t* rec_file;
  const int64_t* ent_rec_ptr;  // [E+1]
  const int64_t* ent_rec_idx;  // [R] record rows grouped by entity
  int32_t* ent_values;         // [E, A] in/out
  const float* theta;          // [A, F]
  const float* phi;            // [Vtot]
  const float* log_phi;        // [Vtot]
  const float* norm_lin;       // [Vtot]
  const float* log_norm;       // [Vtot]
  const int64_t* voff;         // [A+1]
  const int64_t* csr_row_ptr;
  const int32_t* csr_col;
  const float* csr_sim;
  const float* phi_prob;       // [Vtot] alias prob for phi
  const int32_t* phi_alias;    // [Vtot]
  const float* pow_prob;       // power-dist alias tables, concatenated
  const int32_t* pow_alias;
  const int64_t* pow_off;      // [A] offset into pow_* for k=1 (or -1)
  const float* log_pow_total;  // [A * (Kc+1)] log Z_k (index a*(Kc+1)+k)
  const uint8_t* attr_const;
  int Kc;
  int64_t E;
  int A, F;
  int collapsed;
  uint64_t seed;
  uint32_t iteration;
  uint64_t ent_id_base;
  int* error_count;
  // optional [8]: {k2_pairs, hash_pairs, merge_pairs, sum_entries,
  //                sum_units, sum_kobs, rare_pairs, merge_entries}
  unsigned long long* stats;
  int vchunk;  // 1 = chunked hash-accumulate dense path (DBLINK_VCHUNK)
};

// Draw from p(v) ~ phi(v)*norm(v)^k by a dense Gumbel scan (rare path for
// k > Kc where no alias table is cached). Also returns log Z_k via wave sum.
__device__ int dense_power_draw(const ValueArgs& args, int a, int k, uint64_t elem,
                                int lane, float* out_log_total) {
  const int64_t v0 = args.voff[a], v1 = args.voff[a + 1];
  float best = -INFINITY;
  long long best_v = 0;
  double total = 0.0;
  for (int64_t v = v0 + lane; v < v1; v += WAVE) {
    float lw = args.log_phi[v] + (float)k * args.log_norm[v];
    total += exp((double)lw);
    float g = gumbel_from_uniform(philox_uniform(args.seed, args.iteration, PH_VALG,
                                                 elem, (uint32_t)(v - v0) | 0x40000000u));
    if (lw + g > best) { best = lw + g; best_v = v - v0; }
  }
  total = wave_sum(total);
  wave_argmax(best, best_v);
  *out_log_total = (float)log(total);
  return (int)best_v;
}

// Thread-per-pair base draws for empty clusters (k_obs == 0): the base
// distribution is phi for every variant (GibbsUpdates.scala:584-588).
__global__ void value_base_draw_kernel(ValueArgs args) {
  if (args.ctrl != nullptr) {
    args.seed = (uint64_t)args.ctrl[0];
    args.iteration = (uint32_t)args.ctrl[1];
  }
  const int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= args.n_pairs) return;
  const int64_t pair = args.pair_list ? args.pair_list[i] : i;
  if (args.kobs && args.kobs[pair] != 0) return;
  const int64_t e = pair / args.A;
  const int a = (int)(pair % args.A);
  const int64_t v0 = args.voff[a];
  const int V = (int)(args.voff[a + 1] - v0);
  const uint64_t elem = (args.ent_id_base + (uint64_t)e) * 32u + (uint64_t)a;
  float u1, u2;
  philox_uniform2(args.seed, args.iteration, PH_VALM, elem, 0xFFFF0000u, &u1, &u2);
  const int v = alias_draw(args.phi_prob + v0, args.phi_alias + v0, V, u1, u2);
  args.ent_values[e * args.A + a] = (int32_t)v;
}

// Thread-per-pair value update for single-record clusters (k_obs == 1).
// The perturbation distribution for one linked record with value x is STATIC
// per (attribute, x) up to the self-term correction, which is exactly
// (1/theta - 1) of extra raw mass on x. Using the precomputed raw row weights
// raw_w1[j] = phi(col)*norm(col)*(expsim-1), their exclusive row prefix and
// row totals, a draw is: one Philox call, a mixture test, and one binary
// search over the row prefix — O(log row) regardless of row size.
__global__ void value_update_k1_kernel(ValueArgs args) {
  if (args.ctrl != nullptr) {
    args.seed = (uint64_t)args.ctrl[0];
    args.iteration = (uint32_t)args.ctrl[1];
  }
  const int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= args.n_pairs) return;
  const int64_t pair = args.pair_list ? args.pair_list[i] : i;
  if (args.kobs && args.kobs[pair] != 1) return;
  const int64_t e = pair / args.A;
  const int a = (int)(pair % args.A);
  const bool is_const = args.attr_const[a];
  const int64_t v0 = args.voff[a];
  const int V = (int)(args.voff[a + 1] - v0);
  const uint64_t elem = (args.ent_id_base + (uint64_t)e) * 32u + (uint64_t)a;

  // locate the single observed linked record
  const int64_t r_lo = args.ent_rec_ptr[e], r_hi = args.ent_rec_ptr[e + 1];
  int64_t r = -1;
  int32_t x = -1;
  for (int64_t j = r_lo; j < r_hi; ++j) {
    const int64_t rr = args.ent_rec_idx[j];
    const int32_t xx = args.rec_values[rr * args.A + a];
    if (xx >= 0) { r = rr; x = xx; break; }
  }

  // non-collapsed deterministic copy
  if (!args.collapsed && !args.rec_dist[r * args.A + a]) {
    args.ent_values[e * args.A + a] = x;
    return;
  }

  u32x4 rnd = philox4x32(args.seed, (uint32_t)elem, (uint32_t)(elem >> 32),
                         args.iteration ^ (PH_VALM << 24), 0xFFFF0000u);
  const double u_mix = ((double)rnd.x + 0.5) * 2.3283064365386963e-10;
  const double u_sel = ((double)rnd.y + 0.5) * 2.3283064365386963e-10;
  const float u_a1 = u32_to_uniform(rnd.z);
  const float u_a2 = u32_to_uniform(rnd.w);

  auto base_draw = [&]() -> int {
    if (is_const)
      return alias_draw(args.phi_prob + v0, args.phi_alias + v0, V, u_a1, u_a2);
    const int64_t off = args.pow_off[a];  // k = 1 table
    return alias_draw(args.pow_prob + off, args.pow_alias + off, V, u_a1, u_a2);
  };
  if (!args.collapsed && is_const) {
    args.ent_values[e * args.A + a] = (int32_t)base_draw();
    return;
  }

  // raw perturbation mass (in Z-scaled units): row total + self correction
  double extra = 0.0;
  if (args.collapsed) {
    const float th = args.theta[a * args.F + args.rec_file[r]];
    extra = 1.0 / (double)th - 1.0;  // phi*norm*se collapses to (1/theta - 1)
  }
  double raw_total, Wnorm;
  if (is_const) {
    // row is {x}: raw weight = phi(x)*(factor-1) = phi(x)*se = (1/theta-1)
    raw_total = extra;
    Wnorm = extra <= 0.0 ? 0.0 : extra;  // base = phi (Z = 1), raw already phi-scaled
  } else {
    raw_total = args.csr_rawsum[v0 + x] + extra;
    Wnorm = raw_total / args.z1[a];
  }

  int v_new;
  if (u_mix < 1.0 / (1.0 + Wnorm) || raw_total <= 0.0) {
    v_new = base_draw();
  } else if (is_const) {
    v_new = x;  // the only support value
  } else {
    const double t = u_sel * raw_total;
    const int64_t row_lo = args.csr_row_ptr[v0 + x], row_hi = args.csr_row_ptr[v0 + x + 1];
    if (t >= args.csr_rawsum[v0 + x]) {
      v_new = x;  // self-correction mass
    } else {
      // first entry whose exclusive prefix exceeds t, minus one
      int64_t lo = row_lo, hi = row_hi;
      while (lo < hi) {
        const int64_t mid = (lo + hi) >> 1;
        if (args.csr_excl[mid] <= t) lo = mid + 1; else hi = mid;
      }
      int64_t jidx = lo - 1;
      if (jidx < row_lo) jidx = row_lo;
      if (jidx >= row_hi) jidx = row_hi - 1;
      v_new = args.csr_col[jidx];
    }
  }
  args.ent_values[e * args.A + a] = (int32_t)v_new;
}

// Thread-per-pair value update for clusters whose observed linked records
// all share ONE (value, file) — the dominant k >= 2 class (measured d~1.1-1.6
// mean distinct values per pair at 1M/10M stationarity). The perturbation
// weights phi(v) norm(v)^k (e^{k s(x,v)} - 1) are PRECOMPUTED per (value, k)
// (set_value_ktables), so a draw is one Philox call, the theta-dependent
// self-power correction, a mixture test and one binary search — O(log row)
// like the k = 1 kernel, replacing an O(k row) wave merge.
__global__ void value_update_kd1_kernel(ValueArgs args) {
  if (args.ctrl != nullptr) {
    args.seed = (uint64_t)args.ctrl[0];
    args.iteration = (uint32_t)args.ctrl[1];
  }
  const int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= args.n_pairs) return;
  const int64_t pair = i;
  const int k = args.kobs[pair];
  if (k < 2 || k > args.ktab_max) return;
  const int64_t e = pair / args.A;
  const int a = (int)(pair % args.A);
  if (args.attr_const[a]) return;
  const int64_t v0 = args.voff[a];
  const int V = (int)(args.voff[a + 1] - v0);

  // all observed linked records must share one (value, file); any
  // non-distorted observed record (non-collapsed) defers to the wave path
  const int64_t r_lo = args.ent_rec_ptr[e], r_hi = args.ent_rec_ptr[e + 1];
  int32_t x = -1, f = -1;
  for (int64_t j = r_lo; j < r_hi; ++j) {
    const int64_t r = args.ent_rec_idx[j];
    const int32_t xx = args.rec_values[r * args.A + a];
    if (xx < 0) continue;
    if (!args.collapsed && !args.rec_dist[r * args.A + a]) return;
    const int32_t ff = args.collapsed ? args.rec_file[r] : 0;
    if (x < 0) { x = xx; f = ff; }
    else if (xx != x || ff != f) return;  // multi-group: wave path
  }
  if (x < 0) return;
  if (k > args.Kc) return;  // no cached power dist (rare path stays on wave)

  const uint64_t elem = (args.ent_id_base + (uint64_t)e) * 32u + (uint64_t)a;
  u32x4 rnd = philox4x32(args.seed, (uint32_t)elem, (uint32_t)(elem >> 32),
                         args.iteration ^ (PH_VALM << 24), 0xFFFF0000u);
  const double u_mix = ((double)rnd.x + 0.5) * 2.3283064365386963e-10;
  const double u_sel = ((double)rnd.y + 0.5) * 2.3283064365386963e-10;
  const float u_a1 = u32_to_uniform(rnd.z);
  const float u_a2 = u32_to_uniform(rnd.w);

  const int lvl = k - 2;
  // theta-dependent self correction: the table carries the plain self
  // factor e^{k s(x,x)}; collapsed adds ((e^s + se)^k - e^{ks}) phi norm^k
  double extra = 0.0;
  if (args.collapsed) {
    const double th = (double)args.theta[a * args.F + f];
    const double phn = (double)args.phi[v0 + x] * (double)args.norm_lin[v0 + x];
    const double se = (1.0 / th - 1.0) / phn;
    const double es = (double)args.self_expsim[v0 + x];
    double plain = 1.0, boosted = 1.0;
    for (int t = 0; t < k; ++t) { plain *= es; boosted *= es + se; }
    extra = (boosted - plain) *
            exp((double)args.log_phi[v0 + x] + (double)k * (double)args.log_norm[v0 + x]);
  }
  const double row_total = args.tab_rawsum[(int64_t)lvl * (args.voff[args.A]) + v0 + x];
  const double raw_total = row_total + extra;
  const double Zk = exp((double)args.log_pow_total[a * (args.Kc + 1) + k]);
  const double W = raw_total / Zk;

  int v_new;
  if (u_mix < 1.0 / (1.0 + W) || raw_total <= 0.0) {
    const int64_t off = args.pow_off[a] + (int64_t)(k - 1) * V;
    v_new = alias_draw(args.pow_prob + off, args.pow_alias + off, V, u_a1, u_a2);
  } else {
    const double t = u_sel * raw_total;
    if (t >= row_total) {
      v_new = x;  // the self-correction mass
    } else {
      const int64_t row_lo = args.csr_row_ptr[v0 + x];
      const int64_t row_hi = args.csr_row_ptr[v0 + x + 1];
      const double* ex = args.tab_excl + (int64_t)lvl * args.nnz;
      int64_t lo = row_lo, hi = row_hi;
      while (lo < hi) {
        const int64_t mid = (lo + hi) >> 1;
        if (ex[mid] <= t) lo = mid + 1; else hi = mid;
      }
      int64_t jidx = lo - 1;
      if (jidx < row_lo) jidx = row_lo;
      if (jidx >= row_hi) jidx = row_hi - 1;
      v_new = args.csr_col[jidx];
    }
  }
  args.ent_values[e * args.A + a] = (int32_t)v_new;
}

// Wave-per-pair value update for clusters spanning exactly TWO distinct
// values of an attribute (the remaining k >= 2 mass after kd1): the
// perturbation decomposes into two PRECOMPUTED single-row measures
// phi norm^k (e^{m_i s_i} - 1) (set_value_k2tables, level (k-2)(k-1)/2+m-1)
// plus an exact non-negative residual supported on the two self points and
// the row intersection:
//   w(v) = base_k(v)(F1 F2 - 1)
//        = plain1(v) + plain2(v) + base_k(v)(F1-1)(F2-1) [+ self-se boosts]
// Component sampling: two table CDF binary searches, or a residual re-walk.
// The residual walk strides the SHORTER row across the wave's 64 lanes with
// an ordered f64 prefix scan (identical op order in the total and selection
// passes), so it is deterministic and exact.

__device__ void value_update_kd2_pair(const ValueArgs& args, int64_t pair,
                                      int lane) {
  const int k = args.kobs[pair];
  if (k < 2 || k > args.k2tab_max) return;
  const int64_t e = pair / args.A;
  const int a = (int)(pair % args.A);
  if (args.attr_const[a]) return;
  const int64_t v0 = args.voff[a];
  const int V = (int)(args.voff[a + 1] - v0);
  const int64_t Vtot = args.voff[args.A];

  // exactly two first-seen (value, file) groups with DISTINCT values
  // (wave-uniform scalar loads; every lane runs the same scan)
  const int64_t r_lo = args.ent_rec_ptr[e], r_hi = args.ent_rec_ptr[e + 1];
  int32_t x1 = -1, f1 = -1, x2 = -1, f2 = -1;
  int m1 = 0, m2 = 0;
  for (int64_t j = r_lo; j < r_hi; ++j) {
    const int64_t r = args.ent_rec_idx[j];
    const int32_t x = args.rec_values[r * args.A + a];
    if (x < 0) continue;
    if (!args.collapsed && !args.rec_dist[r * args.A + a]) return;
    const int32_t ff = args.collapsed ? args.rec_file[r] : 0;
    if (x1 < 0 || (x == x1 && ff == f1)) { x1 = x; f1 = ff; ++m1; }
    else if (x2 < 0 || (x == x2 && ff == f2)) { x2 = x; f2 = ff; ++m2; }
    else return;  // three or more groups: wave merge path
  }
  if (x2 < 0 || x1 == x2) return;  // d != 2 distinct values

  const int lb = (k - 2) * (k - 1) / 2 - 1;
  const double T1 = args.tab2_rawsum[(int64_t)(lb + m1) * Vtot + v0 + x1];
  const double T2 = args.tab2_rawsum[(int64_t)(lb + m2) * Vtot + v0 + x2];

  auto braw = [&](int32_t v) -> double {
    return exp((double)args.log_phi[v0 + v] +
               (double)k * (double)args.log_norm[v0 + v]);
  };
  auto se_at = [&](int32_t x, int f) -> double {
    if (!args.collapsed) return 0.0;
    const double th = (double)args.theta[a * args.F + f];
    return (1.0 / th - 1.0) /
           ((double)args.phi[v0 + x] * (double)args.norm_lin[v0 + x]);
  };
  const double se1 = se_at(x1, f1), se2 = se_at(x2, f2);

  auto resid_self = [&](int32_t xs, double ses, int ms, int32_t xo,
                        int mo) -> double {
    const double es = (double)args.self_expsim[v0 + xs];
    const float so = sim_lookup(args.csr_row_ptr, args.csr_col, args.csr_sim,
                                v0 + xo, xs);
    const double Fs = pow(es + ses, (double)ms);
    const double Fo = exp((double)so * (double)mo);
    double r = (Fs * Fo - 1.0) - (pow(es, (double)ms) - 1.0);
    if (so != 0.0f) r -= Fo - 1.0;
    r *= braw(xs);
    return r > 0.0 ? r : 0.0;
  };

  const int64_t lo1 = args.csr_row_ptr[v0 + x1], hi1 = args.csr_row_ptr[v0 + x1 + 1];
  const int64_t lo2 = args.csr_row_ptr[v0 + x2], hi2 = args.csr_row_ptr[v0 + x2 + 1];
  const bool first_shorter = (hi1 - lo1) <= (hi2 - lo2);
  const int64_t wlo = first_shorter ? lo1 : lo2;
  const int64_t whi = first_shorter ? hi1 : hi2;
  const int64_t olo = first_shorter ? lo2 : lo1;
  const int64_t ohi = first_shorter ? hi2 : hi1;
  const int mw = first_shorter ? m1 : m2;
  const int mo = first_shorter ? m2 : m1;

  // one lane's residual contribution for walk element j (0 if not in the
  // other row or a self point)
  auto walk_elem = [&](int64_t j) -> double {
    const int32_t v = args.csr_col[j];
    if (v == x1 || v == x2) return 0.0;
    const int64_t p = lower_bound_i32(args.csr_col, olo, ohi, v);
    if (p >= ohi || args.csr_col[p] != v) return 0.0;
    const double Fw = exp((double)args.csr_sim[j] * (double)mw) - 1.0;
    const double Fo = exp((double)args.csr_sim[p] * (double)mo) - 1.0;
    return braw(v) * Fw * Fo;
  };

  const double r_self1 = resid_self(x1, se1, m1, x2, m2);
  const double r_self2 = resid_self(x2, se2, m2, x1, m1);

  // residual pass: lanes stride 64-wide chunks with an ordered inclusive
  // f64 prefix (carry chained across chunks). When target >= 0 the pass
  // selects the first element whose running total crosses it.
  auto residual_pass = [&](double target, int32_t* out_v) -> double {
    double run = r_self1 + r_self2;
    if (out_v != nullptr) {
      if (r_self1 >= target && r_self1 > 0.0) { *out_v = x1; return run; }
      if (run >= target) { *out_v = x2; return run; }
    }
    for (int64_t cb = wlo; cb < whi; cb += WAVE) {
      const int64_t j = cb + lane;
      double mine = (j < whi) ? walk_elem(j) : 0.0;
      // inclusive prefix across lanes (deterministic shfl ladder)
      double pre = mine;
#pragma unroll
      for (int off = 1; off < WAVE; off <<= 1) {
        const double v = __shfl_up(pre, off);
        if (lane >= off) pre += v;
      }
      const double chunk_total = __shfl(pre, WAVE - 1);
      if (out_v != nullptr && run + chunk_total >= target) {
        // first lane whose running total crosses the target
        const bool crossed = (run + pre >= target) && j < whi;
        const unsigned long long mask = __ballot(crossed);
        if (mask != 0ull) {
          const int fl = __ffsll((long long)mask) - 1;
          const int32_t v = args.csr_col[cb + fl];
          *out_v = v;
          return run + chunk_total;
        }
      }
      run += chunk_total;
    }
    return run;
  };
  const double R = residual_pass(-1.0, nullptr);

  const double W = T1 + T2 + R;
  const double Zk = exp((double)args.log_pow_total[a * (args.Kc + 1) + k]);
  const double Wn = W / Zk;

  const uint64_t elem = (args.ent_id_base + (uint64_t)e) * 32u + (uint64_t)a;
  u32x4 rnd = philox4x32(args.seed, (uint32_t)elem, (uint32_t)(elem >> 32),
                         args.iteration ^ (PH_VALM << 24), 0xFFFF0000u);
  const double u_mix = ((double)rnd.x + 0.5) * 2.3283064365386963e-10;
  const double u_sel = ((double)rnd.y + 0.5) * 2.3283064365386963e-10;
  const float u_a1 = u32_to_uniform(rnd.z);
  const float u_a2 = u32_to_uniform(rnd.w);

  int v_new;
  if (u_mix < 1.0 / (1.0 + Wn) || W <= 0.0) {
    const int64_t off = args.pow_off[a] + (int64_t)(k - 1) * V;
    v_new = alias_draw(args.pow_prob + off, args.pow_alias + off, V, u_a1, u_a2);
  } else {
    double t = u_sel * W;
    if (t < T1) {
      const double* ex = args.tab2_excl + (int64_t)(lb + m1) * args.nnz;
      int64_t lo = lo1, hi = hi1;
      while (lo < hi) {
        const int64_t mid = (lo + hi) >> 1;
        if (ex[mid] <= t) lo = mid + 1; else hi = mid;
      }
      int64_t jidx = lo - 1;
      if (jidx < lo1) jidx = lo1;
      if (jidx >= hi1) jidx = hi1 - 1;
      v_new = args.csr_col[jidx];
    } else if (t < T1 + T2) {
      t -= T1;
      const double* ex = args.tab2_excl + (int64_t)(lb + m2) * args.nnz;
      int64_t lo = lo2, hi = hi2;
      while (lo < hi) {
        const int64_t mid = (lo + hi) >> 1;
        if (ex[mid] <= t) lo = mid + 1; else hi = mid;
      }
      int64_t jidx = lo - 1;
      if (jidx < lo2) jidx = lo2;
      if (jidx >= hi2) jidx = hi2 - 1;
      v_new = args.csr_col[jidx];
    } else {
      int32_t sel = x1;
      residual_pass(t - T1 - T2, &sel);
      v_new = sel;
    }
  }
  if (lane == 0) args.ent_values[e * args.A + a] = (int32_t)v_new;
}

__global__ void __launch_bounds__(WAVES_PER_BLOCK_VAL * WAVE)
value_update_kd2_kernel(ValueArgs args) {
  if (args.ctrl != nullptr) {
    args.seed = (uint64_t)args.ctrl[0];
    args.iteration = (uint32_t)args.ctrl[1];
  }
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const int64_t widx = (int64_t)blockIdx.x * WAVES_PER_BLOCK_VAL + wave;
  const int64_t p0 = widx * 8;
  for (int64_t pair = p0; pair < p0 + 8 && pair < args.n_pairs; ++pair) {
    if (args.kobs[pair] >= 2) value_update_kd2_pair(args, pair, lane);
  }
}

constexpr int VAL_STRIDE = 8;  // pairs examined per wave in kobs mode

// VMODE selects which k >= 2 perturbation path a kernel instantiation
// carries: 0 = both (explicit pair lists / tests), 1 = LDS-hash only,
// 2 = union-merge only. The split halves the register pressure of the hot
// kobs-mode kernels (the monolithic kernel sat at 118 VGPRs = 4 waves/SIMD).
template <int VMODE>
__device__ void value_update_pair(const ValueArgs& args, int64_t pair, int lane,
                                  int32_t* keys, float* vals, int32_t* gbuf,
                                  int64_t* glo, int32_t* gn, float* gse);

template <int VMODE>
__global__ void __launch_bounds__(WAVES_PER_BLOCK_VAL * WAVE)
value_update_kernel_t(ValueArgs args) {
  if (args.ctrl != nullptr) {
    args.seed = (uint64_t)args.ctrl[0];
    args.iteration = (uint32_t)args.ctrl[1];
  }
  __shared__ int32_t h_key[WAVES_PER_BLOCK_VAL][HASH_CAP];
  __shared__ float h_val[WAVES_PER_BLOCK_VAL][HASH_CAP];
  __shared__ int32_t g_buf[WAVES_PER_BLOCK_VAL][3 * VAL_DMAX + 1];
  // merge-path per-unit hoists: (row_lo, row_len, se) per distinct group
  __shared__ int64_t g_lo[WAVES_PER_BLOCK_VAL][VAL_DMAX];
  __shared__ int32_t g_n[WAVES_PER_BLOCK_VAL][VAL_DMAX];
  __shared__ float g_se[WAVES_PER_BLOCK_VAL][VAL_DMAX];

  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const int64_t widx = (int64_t)blockIdx.x * WAVES_PER_BLOCK_VAL + wave;
  if (args.pair_list != nullptr) {
    if (widx >= args.n_pairs) return;
    value_update_pair<VMODE>(args, args.pair_list[widx], lane, h_key[wave],
                             h_val[wave], g_buf[wave], g_lo[wave], g_n[wave],
                             g_se[wave]);
  } else {
    // kobs self-selection: each wave examines VAL_STRIDE consecutive pairs
    // and runs the (rare) k >= 2 ones serially
    const int64_t p0 = widx * VAL_STRIDE;
    for (int64_t pair = p0; pair < p0 + VAL_STRIDE && pair < args.n_pairs; ++pair) {
      if (args.kobs[pair] >= 2)
        value_update_pair<VMODE>(args, pair, lane, h_key[wave], h_val[wave],
                                 g_buf[wave], g_lo[wave], g_n[wave], g_se[wave]);
    }
  }
}

template <int VMODE>
__device__ void value_update_pair(const ValueArgs& args, int64_t pair, int lane,
                                  int32_t* keys, float* vals, int32_t* gbuf,
                                  int64_t* glo, int32_t* gn, float* gse) {
  const int64_t e = pair / args.A;
  const int a = (int)(pair % args.A);
  const bool is_const = args.attr_const[a];
  const int64_t v0 = args.voff[a];
  const int V = (int)(args.voff[a + 1] - v0);
  const uint64_t elem = (args.ent_id_base + (uint64_t)e) * 32u + (uint64_t)a;

  // ---- gather observed linked records -------------------------------------
  const int64_t r_lo = args.ent_rec_ptr[e], r_hi = args.ent_rec_ptr[e + 1];
  int k_obs = 0;
  int64_t first_obs_r = -1;
  int32_t first_nondist = -1;  // first observed non-distorted value (record order)
  for (int64_t i = r_lo; i < r_hi; ++i) {
    const int64_t r = args.ent_rec_idx[i];
    const int32_t x = args.rec_values[r * args.A + a];
    if (x < 0) continue;
    if (k_obs == 0) first_obs_r = r;
    ++k_obs;
    if (!args.collapsed && first_nondist < 0 && !args.rec_dist[r * args.A + a])
      first_nondist = x;
  }

  // base distribution: phi for const attrs or k_obs == 0, else power dist k
  auto base_draw = [&](uint32_t draw_tag) -> int {
    float u1, u2;
    philox_uniform2(args.seed, args.iteration, PH_VALM, elem, draw_tag, &u1, &u2);
    if (is_const || k_obs == 0) {
      return alias_draw(args.phi_prob + v0, args.phi_alias + v0, V, u1, u2);
    }
    if (k_obs <= args.Kc) {
      const int64_t off = args.pow_off[a] + (int64_t)(k_obs - 1) * V;
      return alias_draw(args.pow_prob + off, args.pow_alias + off, V, u1, u2);
    }
    float dummy;
    return dense_power_draw(args, a, k_obs, elem, lane, &dummy);
  };

  // non-collapsed deterministic copy (GibbsUpdates.scala:619-630)
  if (!args.collapsed && first_nondist >= 0) {
    if (lane == 0) args.ent_values[e * args.A + a] = first_nondist;
    return;
  }

  if (k_obs == 0) {
    int v = base_draw(0xFFFF0000u);
    if (lane == 0) args.ent_values[e * args.A + a] = (int32_t)v;
    return;
  }

  if (!args.collapsed && is_const) {  // no perturbation for const non-collapsed
    int v = base_draw(0xFFFF0000u);
    if (lane == 0) args.ent_values[e * args.A + a] = (int32_t)v;
    return;
  }

  // log of normalized base probability of value v (local id); log Z for the
  // rare k > Kc case is computed on demand by a wave reduction.
  float log_z_rare = 0.0f;
  if (!is_const && k_obs > args.Kc) {
    double tot = 0.0;
    for (int64_t v = v0 + lane; v < v0 + V; v += WAVE)
      tot += exp((double)(args.log_phi[v] + (float)k_obs * args.log_norm[v]));
    log_z_rare = (float)log(wave_sum(tot));
  }
  const float log_z =
      is_const ? 0.0f
               : (k_obs <= args.Kc ? args.log_pow_total[a * (args.Kc + 1) + k_obs]
                                   : log_z_rare);
  auto log_base_prob = [&](int v_local) -> float {
    const int64_t v = v0 + v_local;
    if (is_const) return args.log_phi[v];
    return args.log_phi[v] + (float)k_obs * args.log_norm[v] - log_z;
  };
  auto self_extra_of = [&](int64_t r, int32_t x) -> float {
    if (!args.collapsed) return 0.0f;
    const float th = args.theta[a * args.F + args.rec_file[r]];
    const float px = args.phi[v0 + x];
    return (1.0f / th - 1.0f) / (is_const ? px : px * args.norm_lin[v0 + x]);
  };
  auto finish = [&](double W, float best, long long best_v) {
    const float u = philox_uniform(args.seed, args.iteration, PH_VALM, elem, 0xFFFF0001u);
    int v_new;
    if ((double)u < 1.0 / (1.0 + W) || best_v < 0) {
      v_new = base_draw(0xFFFF0002u);
    } else {
      v_new = (int)best_v;
    }
    if (lane == 0) args.ent_values[e * args.A + a] = (int32_t)v_new;
  };

  // ---- k == 1 fast path: the perturbation support is one sim row ----------
  if (k_obs == 1) {
    const int64_t r = first_obs_r;
    const int32_t x = args.rec_values[r * args.A + a];
    const float se = self_extra_of(r, x);
    double W = 0.0;
    float best = -INFINITY;
    long long best_v = -1;
    if (is_const) {  // single support value {x}, factor = 1 + se
      if (lane == 0 && se > 0.0f) {
        const float L = __logf(1.0f + se);
        const float logw = log_base_prob(x) + L + __logf(1.0f - __expf(-L));
        W = (double)__expf(logw);
        best = 0.0f;
        best_v = x;
      }
      W = wave_sum(W);
      best_v = __shfl(best_v, 0);
    } else {
      const int64_t row_lo = args.csr_row_ptr[v0 + x], row_hi = args.csr_row_ptr[v0 + x + 1];
      for (int64_t jj = row_lo + lane; jj < row_hi; jj += WAVE) {
        const int32_t v = args.csr_col[jj];
        const float s = args.csr_sim[jj];
        const float L = (v == x && se > 0.0f) ? __logf(__expf(s) + se) : s;
        const float logw = log_base_prob(v) + L + __logf(1.0f - __expf(-L));
        // k = 1 log-weights are bounded (~35), so f32 exp is exact enough and
        // ~20x cheaper than the software double exp
        W += (double)__expf(logw);
        const float g = gumbel_from_uniform(
            philox_uniform(args.seed, args.iteration, PH_VALG, elem, (uint32_t)v));
        if (logw + g > best) { best = logw + g; best_v = v; }
      }
      W = wave_sum(W);
      wave_argmax(best, best_v);
    }
    finish(W, best, best_v);
    return;
  }

  // ---- perturbation weights (k >= 2): grouped by distinct value ----------
  // A cluster's linked records mostly agree, so group them by distinct
  // (value, file): m records sharing x contribute the factor (e^s + se)^m,
  // i.e. one row pass with a multiplier instead of m passes — the union
  // support is sum of DISTINCT rows, not k rows. L_v = sum_g m_g log f_g(v);
  // final weight = base_prob(v) * (exp(L_v) - 1). More than VAL_DMAX
  // distinct groups (rare) falls back to per-record units.
  int d = 0;
  bool gover = false;
  if (lane == 0) {
    for (int64_t i = r_lo; i < r_hi; ++i) {
      const int64_t r = args.ent_rec_idx[i];
      const int32_t x = args.rec_values[r * args.A + a];
      if (x < 0) continue;
      const int32_t f = args.collapsed ? args.rec_file[r] : 0;
      int g = 0;
      for (; g < d; ++g)
        if (gbuf[g] == x && gbuf[VAL_DMAX + g] == f) break;
      if (g < d) {
        gbuf[2 * VAL_DMAX + g] += 1;
      } else if (d < VAL_DMAX) {
        gbuf[d] = x;
        gbuf[VAL_DMAX + d] = f;
        gbuf[2 * VAL_DMAX + d] = 1;
        ++d;
      } else {
        gover = true;
        break;
      }
    }
    gbuf[3 * VAL_DMAX] = gover ? -1 : d;
  }
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_wave_barrier();
  gover = gbuf[3 * VAL_DMAX] < 0;
  d = gover ? 0 : gbuf[3 * VAL_DMAX];

  // self-extra for a group (or record unit): the collapsed (1/theta - 1)
  // correction, constant within a (value, file) group
  auto se_of = [&](int32_t x, int32_t f) -> float {
    if (!args.collapsed) return 0.0f;
    const float th = args.theta[a * args.F + f];
    const float px = args.phi[v0 + x];
    return (1.0f / th - 1.0f) / (is_const ? px : px * args.norm_lin[v0 + x]);
  };
  const int n_units = gover ? k_obs : d;
  // unit u -> (x, multiplicity, file); gover scans the record list
  auto unit_of = [&](int u, int32_t* x, int32_t* m_, int32_t* f) {
    if (!gover) {
      *x = gbuf[u];
      *f = gbuf[VAL_DMAX + u];
      *m_ = gbuf[2 * VAL_DMAX + u];
      return;
    }
    int seen = 0;
    for (int64_t i = r_lo; i < r_hi; ++i) {
      const int64_t r = args.ent_rec_idx[i];
      const int32_t xx = args.rec_values[r * args.A + a];
      if (xx < 0) continue;
      if (seen++ == u) {
        *x = xx;
        *m_ = 1;
        *f = args.collapsed ? args.rec_file[r] : 0;
        return;
      }
    }
  };

  // size the table (and its clear/scan cost) to the actual support
  int64_t total_entries = 0;
  for (int u = 0; u < n_units; ++u) {
    int32_t ux, um, uf;
    unit_of(u, &ux, &um, &uf);
    if (is_const)
      total_entries += 1;
    else
      total_entries += args.csr_row_ptr[v0 + ux + 1] - args.csr_row_ptr[v0 + ux];
  }
  const bool dense = total_entries > (HASH_CAP * 3) / 4;
  if (VMODE == 1 && dense) return;   // the merge kernel's pair
  if (VMODE == 2 && !dense) return;  // the hash kernel's pair
  int tsize = 64;
  while (tsize < 2 * (int)total_entries && tsize < HASH_CAP) tsize <<= 1;
  if (args.stats != nullptr && lane == 0) {
    atomicAdd(&args.stats[0], 1ull);
    atomicAdd(&args.stats[dense ? 2 : 1], 1ull);
    atomicAdd(&args.stats[3], (unsigned long long)total_entries);
    atomicAdd(&args.stats[4], (unsigned long long)n_units);
    atomicAdd(&args.stats[5], (unsigned long long)k_obs);
    if (!is_const && k_obs > args.Kc) atomicAdd(&args.stats[6], 1ull);
    if (dense)
      atomicAdd(&args.stats[7], (unsigned long long)total_entries);
  }

  // single-(value, file) pairs with cached k-tables are the kd1 kernel's
  // (kobs mode only; explicit pair lists keep the full path)
  if (args.kobs != nullptr && args.tab_excl != nullptr && !is_const &&
      !gover && d == 1 && k_obs <= args.ktab_max && k_obs <= args.Kc)
    return;
  // two distinct VALUES with cached (k, m) tables: the kd2 kernel's
  if (args.kobs != nullptr && args.tab2_excl != nullptr && !is_const &&
      !gover && d == 2 && k_obs <= args.k2tab_max && gbuf[0] != gbuf[1])
    return;

  double W = 0.0;            // total perturbation weight
  float best = -INFINITY;    // gumbel-max over perturbation weights
  long long best_v = -1;

  if (VMODE != 2 && !dense) {
    for (int i = lane; i < tsize; i += WAVE) { keys[i] = -1; vals[i] = 0.0f; }
    // drain LDS writes before other lanes' atomics may touch the slots
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_wave_barrier();
    // hash-accumulate log factors per unit
    for (int u = 0; u < n_units; ++u) {
      int32_t ux, um, uf;
      unit_of(u, &ux, &um, &uf);
      const float self_extra = se_of(ux, uf);
      const float fm = (float)um;
      if (is_const) {
        // single-entry row {x}: factor = (1 + self_extra)^m
        if (lane == 0) {
          float logf_ = fm * __logf(1.0f + self_extra);
          uint32_t h = ((uint32_t)ux * 2654435761u) & (tsize - 1);
          while (true) {
            int32_t prev = atomicCAS(&keys[h], -1, ux);
            if (prev == -1 || prev == ux) { atomicAdd(&vals[h], logf_); break; }
            h = (h + 1) & (tsize - 1);
          }
        }
      } else {
        const int64_t row_lo = args.csr_row_ptr[v0 + ux];
        const int64_t row_hi = args.csr_row_ptr[v0 + ux + 1];
        for (int64_t j = row_lo + lane; j < row_hi; j += WAVE) {
          const int32_t v = args.csr_col[j];
          const float s = args.csr_sim[j];  // log expsim > 0
          float factor_log = (v == ux && self_extra > 0.0f)
                                 ? fm * __logf(__expf(s) + self_extra)
                                 : fm * s;
          uint32_t h = ((uint32_t)v * 2654435761u) & (tsize - 1);
          while (true) {
            int32_t prev = atomicCAS(&keys[h], -1, v);
            if (prev == -1 || prev == v) { atomicAdd(&vals[h], factor_log); break; }
            h = (h + 1) & (tsize - 1);
          }
        }
      }
      __builtin_amdgcn_wave_barrier();
    }
    // drain all LDS atomics before cross-lane reads of the table
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_wave_barrier();
    // transform + reduce + gumbel-max over occupied slots
    for (int i = lane; i < tsize; i += WAVE) {
      const int32_t v = keys[i];
      if (v < 0) continue;
      const float L = vals[i];
      // log(exp(L) - 1) = L + log1p(-exp(-L)), stable for L > 0
      const float log_expm1 = L + __logf(1.0f - __expf(-L));
      const float logw = log_base_prob(v) + log_expm1;
      W += (logw < 80.0f) ? (double)__expf(logw) : exp((double)logw);
      const float g = gumbel_from_uniform(
          philox_uniform(args.seed, args.iteration, PH_VALG, elem, (uint32_t)v));
      if (logw + g > best) { best = logw + g; best_v = v; }
    }
  } else if (VMODE != 1 && !gover && !is_const && args.vchunk) {
    // chunked hash-accumulate (union too big for one LDS table): sweep the
    // d rows in ascending value-range chunks sized so each chunk's entries
    // fit the table, then run the same insert/transform/reduce as the small
    // path per chunk. Replaces the union-merge path's (d-1) dependent-load
    // binary searches PER ENTRY with one binary search per unit PER CHUNK.
    // L_v accumulation order per value (ascending unit) matches the small
    // path, so replay stays deterministic. Cursor = row_end - remaining:
    // glo holds the fixed row end, gn the mutable remaining count.
    if (lane == 0) {
      for (int u = 0; u < n_units; ++u) {
        const int32_t x2 = gbuf[u];
        const int64_t lo2 = args.csr_row_ptr[v0 + x2];
        glo[u] = args.csr_row_ptr[v0 + x2 + 1];
        gn[u] = (int32_t)(glo[u] - lo2);
        gse[u] = se_of(x2, gbuf[VAL_DMAX + u]);
      }
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_wave_barrier();
    const int ccap = (HASH_CAP * 3) / 4;
    const int per_u = ccap / n_units > 0 ? ccap / n_units : 1;
    for (;;) {
      // chunk bound (exclusive): min over units of the value per_u entries
      // ahead of the cursor; every per-unit segment is then <= per_u
      int32_t vhi = INT32_MAX;
      bool any = false;
      for (int u = 0; u < n_units; ++u) {
        const int64_t end2 = glo[u];
        const int64_t cu = end2 - gn[u];
        if (cu < end2) any = true;
        if (cu + per_u < end2) {
          const int32_t cv = args.csr_col[cu + per_u];
          if (cv < vhi) vhi = cv;
        }
      }
      if (!any) break;
      for (int i = lane; i < HASH_CAP; i += WAVE) { keys[i] = -1; vals[i] = 0.0f; }
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_wave_barrier();
      for (int u = 0; u < n_units; ++u) {
        const int64_t end2 = glo[u];
        const int64_t cu = end2 - gn[u];
        const int64_t seg_end =
            (vhi == INT32_MAX) ? end2
                               : lower_bound_i32(args.csr_col, cu, end2, vhi);
        const int32_t ux = gbuf[u];
        const float self_extra = gse[u];
        const float fm = (float)gbuf[2 * VAL_DMAX + u];
        for (int64_t j = cu + lane; j < seg_end; j += WAVE) {
          const int32_t v = args.csr_col[j];
          const float s = args.csr_sim[j];
          const float factor_log = (v == ux && self_extra > 0.0f)
                                       ? fm * __logf(__expf(s) + self_extra)
                                       : fm * s;
          uint32_t h = ((uint32_t)v * 2654435761u) & (HASH_CAP - 1);
          while (true) {
            const int32_t prev = atomicCAS(&keys[h], -1, v);
            if (prev == -1 || prev == v) { atomicAdd(&vals[h], factor_log); break; }
            h = (h + 1) & (HASH_CAP - 1);
          }
        }
        if (lane == 0) gn[u] = (int32_t)(end2 - seg_end);
        __builtin_amdgcn_wave_barrier();
      }
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_wave_barrier();
      for (int i = lane; i < HASH_CAP; i += WAVE) {
        const int32_t v = keys[i];
        if (v < 0) continue;
        const float L = vals[i];
        const float log_expm1 = L + __logf(1.0f - __expf(-L));
        const float logw = log_base_prob(v) + log_expm1;
        W += (logw < 80.0f) ? (double)__expf(logw) : exp((double)logw);
        const float g = gumbel_from_uniform(
            philox_uniform(args.seed, args.iteration, PH_VALG, elem, (uint32_t)v));
        if (logw + g > best) { best = logw + g; best_v = v; }
      }
      // reads of this chunk's table must drain before the next chunk's clear
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_wave_barrier();
    }
  } else if (VMODE != 1) {
    // union-merge path (gover / const attrs): iterate every entry of every
    // DISTINCT row, but process a value only from the FIRST unit containing
    // it; full L_v comes from binary searches in the other units' rows.
    // Per-unit metadata (row bounds, the theta-dependent self term) is
    // hoisted into LDS once per pair — the entry loop touches ~10^2-10^3
    // values per unit and must not recompute it per (entry, unit).
    const bool hoist = !gover && !is_const;
    if (hoist && lane == 0) {
      for (int u = 0; u < n_units; ++u) {
        const int32_t x2 = gbuf[u];
        glo[u] = args.csr_row_ptr[v0 + x2];
        gn[u] = (int32_t)(args.csr_row_ptr[v0 + x2 + 1] - glo[u]);
        gse[u] = se_of(x2, gbuf[VAL_DMAX + u]);
      }
    }
    for (int u = 0; u < n_units; ++u) {
      int32_t ux, um, uf;
      unit_of(u, &ux, &um, &uf);
      if (is_const) {
        if (lane == 0) {
          bool first = true;
          for (int u2 = 0; u2 < u && first; ++u2) {
            int32_t x2, m2, f2;
            unit_of(u2, &x2, &m2, &f2);
            if (x2 == ux) first = false;  // only possible when gover
          }
          if (first) {
            float L = 0.0f;
            for (int u2 = u; u2 < n_units; ++u2) {
              int32_t x2, m2, f2;
              unit_of(u2, &x2, &m2, &f2);
              if (x2 != ux) continue;
              L += (float)m2 * __logf(1.0f + se_of(x2, f2));
            }
            if (L > 0.0f) {
              const float log_expm1 = L + __logf(1.0f - __expf(-L));
              const float logw = log_base_prob(ux) + log_expm1;
              W += (logw < 80.0f) ? (double)__expf(logw) : exp((double)logw);
              const float g = gumbel_from_uniform(philox_uniform(
                  args.seed, args.iteration, PH_VALG, elem, (uint32_t)ux));
              if (logw + g > best) { best = logw + g; best_v = ux; }
            }
          }
        }
        continue;
      }
      const int64_t row_lo = args.csr_row_ptr[v0 + ux];
      const int64_t row_hi = args.csr_row_ptr[v0 + ux + 1];
      for (int64_t j = row_lo + lane; j < row_hi; j += WAVE) {
        const int32_t v = args.csr_col[j];
        // one ascending pass fuses the dedupe test (an EARLIER unit's row
        // containing v means that unit owns the value) with the log-factor
        // accumulation: one binary search per (entry, unit) instead of two
        bool first = true;
        float L = 0.0f;
        for (int u2 = 0; u2 < n_units; ++u2) {
          int32_t x2, m2;
          float se2;
          float s2;
          if (hoist) {
            x2 = gbuf[u2];
            m2 = gbuf[2 * VAL_DMAX + u2];
            se2 = gse[u2];
            if (u2 == u) {
              s2 = args.csr_sim[j];
            } else {
              const int64_t lo2 = glo[u2];
              const int64_t hi2 = lo2 + gn[u2];
              const int64_t p2 = lower_bound_i32(args.csr_col, lo2, hi2, v);
              const bool found = p2 < hi2 && args.csr_col[p2] == v;
              if (found && u2 < u) { first = false; break; }
              s2 = found ? args.csr_sim[p2] : 0.0f;
            }
          } else {
            int32_t f2;
            unit_of(u2, &x2, &m2, &f2);
            if (u2 == u) {
              s2 = args.csr_sim[j];
            } else {
              const int64_t lo2 = args.csr_row_ptr[v0 + x2];
              const int64_t hi2 = args.csr_row_ptr[v0 + x2 + 1];
              const int64_t p2 = lower_bound_i32(args.csr_col, lo2, hi2, v);
              const bool found = p2 < hi2 && args.csr_col[p2] == v;
              if (found && u2 < u) { first = false; break; }
              s2 = found ? args.csr_sim[p2] : 0.0f;
            }
            se2 = se_of(x2, f2);
          }
          if (v == x2 && se2 > 0.0f)
            L += (float)m2 * __logf(__expf(s2) + se2);
          else if (s2 != 0.0f)
            L += (float)m2 * s2;
        }
        if (!first) continue;
        const float log_expm1 = L + __logf(1.0f - __expf(-L));
        const float logw = log_base_prob(v) + log_expm1;
        W += (logw < 80.0f) ? (double)__expf(logw) : exp((double)logw);
        const float g = gumbel_from_uniform(
            philox_uniform(args.seed, args.iteration, PH_VALG, elem, (uint32_t)v));
        if (logw + g > best) { best = logw + g; best_v = v; }
      }
    }
  }
  W = wave_sum(W);
  wave_argmax(best, best_v);

  // mixture between base and perturbation (GibbsUpdates.scala:593-597)
  finish(W, best, best_v);
}

// Brute-force dense value update (Gibbs-Sequential, GibbsUpdates.scala:652-698)
__global__ void value_update_seq_kernel(ValueArgs args) {
  if (args.ctrl != nullptr) {
    args.seed = (uint64_t)args.ctrl[0];
    args.iteration = (uint32_t)args.ctrl[1];
  }
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const int64_t pair = (int64_t)blockIdx.x * (blockDim.x / WAVE) + wave;
  if (pair >= args.n_pairs) return;
  const int64_t e = pair / args.A;
  const int a = (int)(pair % args.A);
  const bool is_const = args.attr_const[a];
  const int64_t v0 = args.voff[a];
  const int V = (int)(args.voff[a + 1] - v0);
  const uint64_t elem = (args.ent_id_base + (uint64_t)e) * 32u + (uint64_t)a;

  const int64_t r_lo = args.ent_rec_ptr[e], r_hi = args.ent_rec_ptr[e + 1];
  int k_obs = 0;
  int32_t first_nondist = -1;
  for (int64_t i = r_lo; i < r_hi; ++i) {
    const int64_t r = args.ent_rec_idx[i];
    const int32_t x = args.rec_values[r * args.A + a];
    if (x < 0) continue;
    ++k_obs;
    if (first_nondist < 0 && !args.rec_dist[r * args.A + a]) first_nondist = x;
  }
  if (first_nondist >= 0 && k_obs > 0) {
    if (lane == 0) args.ent_values[e * args.A + a] = first_nondist;
    return;
  }
  if (k_obs == 0 || is_const) {
    float u1, u2;
    philox_uniform2(args.seed, args.iteration, PH_VALM, elem, 0xFFFF0000u, &u1, &u2);
    int v = alias_draw(args.phi_prob + v0, args.phi_alias + v0, V, u1, u2);
    if (lane == 0) args.ent_values[e * args.A + a] = (int32_t)v;
    return;
  }
  // dense scan: w(v) = phi(v) * prod_r [expsim(x_r, v) * norm(v) * phi(x_r)]
  float best = -INFINITY;
  long long best_v = 0;
  for (int v_local = lane; v_local < V; v_local += WAVE) {
    float lw = args.log_phi[v0 + v_local];
    for (int64_t i = r_lo; i < r_hi; ++i) {
      const int64_t r = args.ent_rec_idx[i];
      const int32_t x = args.rec_values[r * args.A + a];
      if (x < 0) continue;
      lw += sim_lookup(args.csr_row_ptr, args.csr_col, args.csr_sim, v0 + x, v_local) +
            args.log_norm[v0 + v_local] + args.log_phi[v0 + x];
    }
    const float g = gumbel_from_uniform(
        philox_uniform(args.seed, args.iteration, PH_VALG, elem, (uint32_t)v_local));
    if (lw + g > best) { best = lw + g; best_v = v_local; }
  }
  wave_argmax(best, best_v);
  if (lane == 0) args.ent_values[e * args.A + a] = (int32_t)best_v;
}

// ---------------------------------------------------------------------------
// K7 (+K8): distortion update, with the log-likelihood reduction fused in.
// The distorted-record likelihood terms depend on exactly the data this
// kernel just touched (x, y, z), and the entity prior terms ride along as
// extra leading indices — fusing removes a second full pass over the state
// (summary_loglik_kernel remains for the standalone initial summary).
// ---------------------------------------------------------------------------

__global__ void distortion_update_kernel(
    const int32_t* __restrict__ rec_values, uint8_t* __restrict__ rec_dist,
    const int32_t* __restrict__ rec_file, const int64_t* __restrict__ rec_gid,
    const int64_t* __restrict__ rec_ent, const int32_t* __restrict__ ent_values,
    const float* __restrict__ theta, const float* __restrict__ phi,
    const float* __restrict__ norm_lin, const float* __restrict__ self_expsim,
    const int64_t* __restrict__ voff, const uint8_t* __restrict__ attr_const,
    int64_t R, int A, int F, uint64_t seed, uint32_t iteration,
    const int64_t* __restrict__ ctrl,
    const float* __restrict__ log_phi, const float* __restrict__ log_norm,
    const int64_t* __restrict__ csr_row_ptr, const int32_t* __restrict__ csr_col,
    const float* __restrict__ csr_sim,
    int64_t E,                      // entity prior rows (0 when not fusing)
    double* __restrict__ loglik) {  // nullptr = distortion only
  const int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (ctrl != nullptr) { seed = (uint64_t)ctrl[0]; iteration = (uint32_t)ctrl[1]; }
  double contrib = 0.0;
  if (idx < E * A) {
    const int64_t e = idx / A;
    const int a = (int)(idx % A);
    contrib = (double)log_phi[voff[a] + ent_values[e * A + a]];
  } else if (idx < (E + R) * A) {
    const int64_t j = idx - E * A;
    const int64_t r = j / A;
    const int a = (int)(j % A);
    const int32_t x = rec_values[j];
    const float th = theta[a * F + rec_file[r]];
    const float u = philox_uniform(seed, iteration, PH_DIST,
                                   (uint64_t)rec_gid[r] * 32u + (uint64_t)a, 0);
    uint8_t z;
    if (x < 0) {
      z = u < th;
    } else {
      const int32_t y = ent_values[rec_ent[r] * A + a];
      if (x == y) {
        float pr1 = th * phi[voff[a] + x];
        if (!attr_const[a]) pr1 *= norm_lin[voff[a] + x] * self_expsim[voff[a] + x];
        const float pr0 = 1.0f - th;
        const float psum = pr0 + pr1;
        const float pz = psum != 0.0f ? pr1 / psum : 0.0f;
        z = u < pz;
      } else {
        z = 1;
      }
    }
    rec_dist[j] = z;
    if (loglik != nullptr && z && x >= 0) {
      float lp = log_phi[voff[a] + x];
      if (!attr_const[a]) {
        const int32_t y = ent_values[rec_ent[r] * A + a];
        lp += log_norm[voff[a] + y] +
              sim_lookup(csr_row_ptr, csr_col, csr_sim, voff[a] + x, y);
      }
      contrib = (double)lp;
    }
  }
  if (loglik == nullptr) return;
  // wave shuffle reduce, then cross-wave LDS, one atomic per block
  for (int off = WAVE / 2; off > 0; off >>= 1)
    contrib += __shfl_down(contrib, off, WAVE);
  __shared__ double wsum[256 / WAVE];
  const int wid = threadIdx.x / WAVE;
  if ((threadIdx.x & (WAVE - 1)) == 0) wsum[wid] = contrib;
  __syncthreads();
  if (threadIdx.x == 0) {
    double s = 0.0;
    const int nw = (int)(blockDim.x / WAVE);
    for (int i = 0; i < nw; ++i) s += wsum[i];
    // spread over 256 slots: every block adding one global f64 serializes
    // (~10 ns per same-address atomic x tens of thousands of blocks)
    if (s != 0.0) atomicAdd(&loglik[blockIdx.x & 255], s);
  }
}

// ---------------------------------------------------------------------------
// K8: log-likelihood reduction (entity priors + distorted record terms)
// ---------------------------------------------------------------------------

__global__ void summary_loglik_kernel(
    const int32_t* __restrict__ ent_values, const int32_t* __restrict__ rec_values,
    const uint8_t* __restrict__ rec_dist, const int64_t* __restrict__ rec_ent,
    const float* __restrict__ log_phi, const float* __restrict__ log_norm,
    const int64_t* __restrict__ voff, const int64_t* __restrict__ csr_row_ptr,
    const int32_t* __restrict__ csr_col, const float* __restrict__ csr_sim,
    const uint8_t* __restrict__ attr_const, int64_t E, int64_t R, int A,
    double* __restrict__ out) {
  const int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t total = (E + R) * A;
  double contrib = 0.0;
  if (idx < E * A) {
    const int64_t e = idx / A;
    const int a = (int)(idx % A);
    contrib = (double)log_phi[voff[a] + ent_values[e * A + a]];
  } else if (idx < total) {
    const int64_t j = idx - E * A;
    const int64_t r = j / A;
    const int a = (int)(j % A);
    if (rec_dist[r * A + a]) {
      const int32_t x = rec_values[r * A + a];
      if (x >= 0) {
        float lp = log_phi[voff[a] + x];
        if (!attr_const[a]) {
          const int32_t y = ent_values[rec_ent[r] * A + a];
          lp += log_norm[voff[a] + y] +
                sim_lookup(csr_row_ptr, csr_col, csr_sim, voff[a] + x, y);
        }
        contrib = (double)lp;
      }
    }
  }
  // block reduce then one atomic per block
  __shared__ double partial[256];
  partial[threadIdx.x] = contrib;
  __syncthreads();
  for (int s = blockDim.x / 2; s > 0; s >>= 1) {
    if (threadIdx.x < s) partial[threadIdx.x] += partial[threadIdx.x + s];
    __syncthreads();
  }
  if (threadIdx.x == 0) atomicAdd(&out[blockIdx.x & 255], partial[0]);
}

// ---------------------------------------------------------------------------
// Fused support kernels (small-problem launch-count reduction)
// ---------------------------------------------------------------------------

// Build inverted-index sort keys for entities and query keys for records in
// one launch: key = (part * A + a) * Vmax + value (a-major for entities).
__global__ void build_keys_kernel(
    const int32_t* __restrict__ ent_part, const int32_t* __restrict__ ent_values,
    const int32_t* __restrict__ rec_part, const int32_t* __restrict__ rec_values,
    const int32_t* __restrict__ pair_a1,   // [NP] first attr of pseudo slot
    const int32_t* __restrict__ pair_a2,   // [NP] second attr
    const int32_t* __restrict__ pair_v2,   // [NP] domain size of second attr
    int64_t E, int64_t R, int A, int NP, int64_t Vmax,
    int64_t* __restrict__ ekeys,   // [(A+NP) * E] slot-major
    int64_t* __restrict__ qkeys) { // [R * (A+NP)]
  const int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int T = A + NP;
  if (idx < E * T) {
    const int t = (int)(idx / E);
    const int64_t e = idx % E;
    int64_t v;
    if (t < A) {
      v = ent_values[e * A + t];
    } else {
      // composite value of a constant-attribute pair
      v = (int64_t)ent_values[e * A + pair_a1[t - A]] * pair_v2[t - A]
          + ent_values[e * A + pair_a2[t - A]];
    }
    ekeys[idx] = ((int64_t)ent_part[e] * T + t) * Vmax + v;
  } else if (idx < E * T + R * T) {
    const int64_t j = idx - E * T;
    const int64_t r = j / T;
    const int t = (int)(j % T);
    int64_t v;
    if (t < A) {
      const int32_t x = rec_values[r * A + t];
      v = x < 0 ? 0 : x;
    } else {
      const int32_t x1 = rec_values[r * A + pair_a1[t - A]];
      const int32_t x2 = rec_values[r * A + pair_a2[t - A]];
      v = (x1 < 0 || x2 < 0) ? 0 : (int64_t)x1 * pair_v2[t - A] + x2;
    }
    qkeys[j] = ((int64_t)rec_part[r] * T + t) * Vmax + v;
  }
}

// ---------------------------------------------------------------------------
// Counting-sort inverted index (replaces radix sort + batched searchsorted):
//   1. postings_hist_kernel:    counts[key] += 1 over all (entity, slot)
//   2. exclusive prefix (torch.cumsum, host-issued) -> dense ptr over keys
//   3. postings_scatter_kernel: postings[cursor[key]++] = entity
//   4. cand_ranges_kernel:      per (record, slot) lo = ptr[key], hi = ptr[key+1]
// Posting order within one key is arbitrary (atomic cursors): the link
// kernels only ENUMERATE a posting range — membership tests are direct
// entity-value comparisons and the Gumbel-max draw is keyed by entity id,
// so the sampled distribution is order-independent.
// Key layout matches build_keys_kernel: (part * T + t) * Vmax + v.
// ---------------------------------------------------------------------------

DBL_D int64_t posting_slot_value(const int32_t* vals, int64_t row, int A, int t,
                                 const int32_t* pair_a1, const int32_t* pair_a2,
                                 const int32_t* pair_v2) {
  if (t < A) return vals[row * A + t];
  return (int64_t)vals[row * A + pair_a1[t - A]] * pair_v2[t - A]
         + vals[row * A + pair_a2[t - A]];
}

// Popular values (low-cardinality constant attributes) would serialize tens
// of thousands of same-address global atomics, so both passes pre-aggregate
// duplicate keys in an LDS open-addressing hash per block: one global atomic
// per DISTINCT key per block. Keys fit int32 (the dense path is capped at
// 2^28 counters), so the table uses 32-bit LDS CAS.
constexpr int PB_THREADS = 1024;  // items per block == threads
constexpr int PB_TBL = 2048;      // >= 2x max distinct keys per block

DBL_D int pb_hash(int32_t key) {
  return (int)(((uint32_t)key * 2654435761u) >> 16) & (PB_TBL - 1);
}

__global__ __launch_bounds__(PB_THREADS) void postings_hist_kernel(
    const int32_t* __restrict__ ent_part, const int32_t* __restrict__ ent_values,
    const int32_t* __restrict__ pair_a1, const int32_t* __restrict__ pair_a2,
    const int32_t* __restrict__ pair_v2, int64_t E, int A, int NP, int64_t Vmax,
    int32_t* __restrict__ counts) {
  __shared__ int32_t h_key[PB_TBL];
  __shared__ int32_t h_cnt[PB_TBL];
  for (int i = threadIdx.x; i < PB_TBL; i += blockDim.x) {
    h_key[i] = -1;
    h_cnt[i] = 0;
  }
  __syncthreads();
  const int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int T = A + NP;
  if (idx < E * T) {
    const int t = (int)(idx / E);
    const int64_t e = idx % E;
    const int64_t v = posting_slot_value(ent_values, e, A, t, pair_a1, pair_a2, pair_v2);
    const int32_t key = (int32_t)(((int64_t)ent_part[e] * T + t) * Vmax + v);
    int slot = pb_hash(key);
    for (;;) {
      const int32_t prev = atomicCAS(&h_key[slot], -1, key);
      if (prev == -1 || prev == key) {
        atomicAdd(&h_cnt[slot], 1);
        break;
      }
      slot = (slot + 1) & (PB_TBL - 1);
    }
  }
  __syncthreads();
  for (int i = threadIdx.x; i < PB_TBL; i += blockDim.x)
    if (h_key[i] >= 0) atomicAdd(&counts[h_key[i]], h_cnt[i]);
}

__global__ __launch_bounds__(PB_THREADS) void postings_scatter_kernel(
    const int32_t* __restrict__ ent_part, const int32_t* __restrict__ ent_values,
    const int32_t* __restrict__ pair_a1, const int32_t* __restrict__ pair_a2,
    const int32_t* __restrict__ pair_v2, int64_t E, int A, int NP, int64_t Vmax,
    int32_t* __restrict__ cursor,        // [NK] initialized to exclusive prefix
    int32_t* __restrict__ postings) {    // [E * T]
  __shared__ int32_t h_key[PB_TBL];
  __shared__ int32_t h_cnt[PB_TBL];
  __shared__ int32_t h_base[PB_TBL];
  for (int i = threadIdx.x; i < PB_TBL; i += blockDim.x) {
    h_key[i] = -1;
    h_cnt[i] = 0;
  }
  __syncthreads();
  const int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int T = A + NP;
  int my_slot = -1;
  int32_t my_rank = 0;
  int32_t e_out = 0;
  if (idx < E * T) {
    const int t = (int)(idx / E);
    const int64_t e = idx % E;
    e_out = (int32_t)e;
    const int64_t v = posting_slot_value(ent_values, e, A, t, pair_a1, pair_a2, pair_v2);
    const int32_t key = (int32_t)(((int64_t)ent_part[e] * T + t) * Vmax + v);
    int slot = pb_hash(key);
    for (;;) {
      const int32_t prev = atomicCAS(&h_key[slot], -1, key);
      if (prev == -1 || prev == key) {
        my_rank = atomicAdd(&h_cnt[slot], 1);
        my_slot = slot;
        break;
      }
      slot = (slot + 1) & (PB_TBL - 1);
    }
  }
  __syncthreads();
  for (int i = threadIdx.x; i < PB_TBL; i += blockDim.x)
    if (h_key[i] >= 0) h_base[i] = atomicAdd(&cursor[h_key[i]], h_cnt[i]);
  __syncthreads();
  if (my_slot >= 0) postings[h_base[my_slot] + my_rank] = e_out;
}

// One thread per record: route each record to one of three link paths.
//   mode 0 = wave scan (link_update_kernel)
//   mode 1 = thread scan (link_update_small_kernel, short candidate list)
//   mode 2 = hierarchical sampler (link_update_heavy_kernel): records whose
//            candidate set is a large partition slice — no observed
//            non-distorted attribute (pool = whole partition), or exactly one
//            with a big posting range (pool = that stable posting segment) —
//            where an exact scan would touch thousands of entities.
__global__ void classify_modes_kernel(
    const int32_t* __restrict__ rec_values, const uint8_t* __restrict__ rec_dist,
    const int32_t* __restrict__ rec_part, const int64_t* __restrict__ ent_ptr,
    const int64_t* __restrict__ cand_lo, const int64_t* __restrict__ cand_hi,
    int64_t R, int A, int NP, int64_t small_threshold, int64_t heavy_threshold,
    uint8_t* __restrict__ mode) {
  const int64_t r = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (r >= R) return;
  const int T = A + NP;
  int64_t best = INT64_MAX;
  int nd_count = 0;
  for (int a = 0; a < A; ++a) {
    if (rec_values[r * A + a] < 0 || rec_dist[r * A + a]) continue;
    ++nd_count;
    const int64_t n = cand_hi[r * T + a] - cand_lo[r * T + a];
    if (n < best) best = n;
  }
  uint8_t m = 0;
  if (heavy_threshold > 0) {
    const int32_t p = rec_part[r];
    const int64_t pool = ent_ptr[p + 1] - ent_ptr[p];
    if (nd_count == 0 && pool > heavy_threshold) m = 2;
    else if (nd_count == 1 && best > heavy_threshold) m = 2;
  }
  if (m == 0 && best != INT64_MAX && best <= small_threshold) m = 1;
  mode[r] = m;
}

__global__ void cand_ranges_kernel(
    const int32_t* __restrict__ rec_part, const int32_t* __restrict__ rec_values,
    const int32_t* __restrict__ pair_a1, const int32_t* __restrict__ pair_a2,
    const int32_t* __restrict__ pair_v2, const int64_t* __restrict__ ptr,
    int64_t R, int A, int NP, int64_t Vmax,
    int64_t* __restrict__ cand_lo, int64_t* __restrict__ cand_hi) {  // [R * T]
  const int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int T = A + NP;
  if (idx >= R * T) return;
  const int64_t r = idx / T;
  const int t = (int)(idx % T);
  int64_t v;
  if (t < A) {
    v = rec_values[r * A + t];
  } else {
    const int32_t x1 = rec_values[r * A + pair_a1[t - A]];
    const int32_t x2 = rec_values[r * A + pair_a2[t - A]];
    v = (x1 < 0 || x2 < 0) ? -1 : (int64_t)x1 * pair_v2[t - A] + x2;
  }
  if (v < 0) {  // missing value: empty range (never consulted as a constraint)
    cand_lo[idx] = 0;
    cand_hi[idx] = 0;
    return;
  }
  const int64_t key = ((int64_t)rec_part[r] * T + t) * Vmax + v;
  cand_lo[idx] = ptr[key];
  cand_hi[idx] = ptr[key + 1];
}

// Summary counts in one pass: per-record distortion histogram + per
// (attr, file) aggregates + isolate count (from the entity->record CSR).
__global__ void summary_counts_kernel(
    const uint8_t* __restrict__ rec_dist, const int32_t* __restrict__ rec_file,
    const int64_t* __restrict__ ent_rec_ptr, int64_t E, int64_t R, int A, int F,
    int n_counts,
    unsigned long long* __restrict__ counts) {  // [1 + A*F + A+1]: iso, agg, hist
  // stage all counters in LDS: one global atomic per counter per BLOCK
  // (per-element global atomics on ~12 counters serialize at ~11 ns each)
  extern __shared__ unsigned int block_counts[];
  for (int i = threadIdx.x; i < n_counts; i += blockDim.x) block_counts[i] = 0u;
  __syncthreads();

  const int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx < E) {
    if (ent_rec_ptr[idx + 1] == ent_rec_ptr[idx]) atomicAdd(&block_counts[0], 1u);
  } else if (idx < E + R) {
    const int64_t r = idx - E;
    int nd = 0;
    const int f = rec_file[r];
    for (int a = 0; a < A; ++a) {
      if (rec_dist[r * A + a]) {
        ++nd;
        atomicAdd(&block_counts[1 + a * F + f], 1u);
      }
    }
    atomicAdd(&block_counts[1 + A * F + nd], 1u);
  }
  __syncthreads();
  for (int i = threadIdx.x; i < n_counts; i += blockDim.x) {
    if (block_counts[i]) atomicAdd(&counts[i], (unsigned long long)block_counts[i]);
  }
}

// Pack loglik + counts into the f64 summary buffer (one tiny launch).
__global__ void summary_finalize_kernel(
    const double* __restrict__ loglik, const unsigned long long* __restrict__ counts,
    int n_counts, double* __restrict__ packed) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i == 0) {
    double s = 0.0;
    for (int j = 0; j < 256; ++j) s += loglik[j];
    packed[0] = s;
  }
  if (i < n_counts) packed[1 + i] = (double)counts[i];
}

// ---------------------------------------------------------------------------
// MFMA experiment: the most matrix-shaped hot computation is
// the dense categorical-agreement block of the PCG-II link weights —
// score[r, e] = sum_a bonus_a(x_{r,a}) * 1{x_{r,a} == y_{e,a}} over the
// constant attributes. Two implementations of the same R x E scorer:
//   scalar_score_bench: LDS-staged entity codes, wave-per-record compare/add
//   mfma_score_bench:   one-hot formulation U[r,k] * H[k,e] on the bf16
//                       matrix cores (mfma_f32_16x16x32_bf16, K padded to a
//                       multiple of 32), fragments synthesized on the fly
//                       from the compact codes (no dense one-hot in memory)
// The measured comparison lives in profiles/README.md.
// ---------------------------------------------------------------------------

typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

DBL_D short f32_to_bf16(float f) {
  union { float f; uint32_t u; } c = {f};
  return (short)(c.u >> 16);  // truncate (bench precision is bf16 anyway)
}

// score tile per WAVE: 16 records x 16 entities, K-loop over one-hot domain
__global__ void mfma_score_kernel(
    const int32_t* __restrict__ rcode,   // [R, 3] concatenated-domain codes
    const float* __restrict__ rbonus,    // [R, 3] per-record attr bonuses
    const int32_t* __restrict__ ecode,   // [E, 3]
    int R, int E, int K,                 // K = padded one-hot width
    float* __restrict__ score) {         // [R, E]
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const int tiles_e = E / 16;
  const int tile = blockIdx.x * (blockDim.x / WAVE) + wave;
  const int tr = tile / tiles_e, te = tile % tiles_e;
  if (tr * 16 >= R) return;

  // A: row i = lane & 15 (record), 8 elements at k = k0 + (lane>>4)*8 + j
  // B: col j = lane & 15 (entity), same k layout
  // C/D: col = lane & 15, row = (lane>>4)*4 + reg
  const int r = tr * 16 + (lane & 15);
  const int e = te * 16 + (lane & 15);
  int32_t rc0 = rcode[r * 3], rc1 = rcode[r * 3 + 1], rc2 = rcode[r * 3 + 2];
  float rb0 = rbonus[r * 3], rb1 = rbonus[r * 3 + 1], rb2 = rbonus[r * 3 + 2];
  int32_t ec0 = ecode[e * 3], ec1 = ecode[e * 3 + 1], ec2 = ecode[e * 3 + 2];

  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  for (int k0 = 0; k0 < K; k0 += 32) {
    bf16x8 fa, fb;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int k = k0 + (lane >> 4) * 8 + j;
      float av = (k == rc0) ? rb0 : (k == rc1) ? rb1 : (k == rc2) ? rb2 : 0.0f;
      fa[j] = f32_to_bf16(av);
      fb[j] = f32_to_bf16((k == ec0 || k == ec1 || k == ec2) ? 1.0f : 0.0f);
    }
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(fa, fb, acc, 0, 0, 0);
  }
#pragma unroll
  for (int reg = 0; reg < 4; ++reg) {
    const int row = (lane >> 4) * 4 + reg;
    score[(int64_t)(tr * 16 + row) * E + te * 16 + (lane & 15)] = acc[reg];
  }
}

// Wave-per-record scalar scorer over LDS-staged packed entity codes.
constexpr int SCALAR_ETILE = 1024;

__global__ void scalar_score_kernel(
    const int32_t* __restrict__ rcode, const float* __restrict__ rbonus,
    const int32_t* __restrict__ ecode, int R, int E,
    float* __restrict__ score) {
  __shared__ uint32_t epack[SCALAR_ETILE];
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const int r = blockIdx.x * (blockDim.x / WAVE) + wave;
  const uint32_t rc = r < R ? (uint32_t)(rcode[r * 3] | (rcode[r * 3 + 1] << 10)
                                         | (rcode[r * 3 + 2] << 20)) : 0u;
  const float b0 = r < R ? rbonus[r * 3] : 0.f;
  const float b1 = r < R ? rbonus[r * 3 + 1] : 0.f;
  const float b2 = r < R ? rbonus[r * 3 + 2] : 0.f;
  for (int e0 = 0; e0 < E; e0 += SCALAR_ETILE) {
    __syncthreads();
    for (int i = threadIdx.x; i < SCALAR_ETILE && e0 + i < E; i += blockDim.x)
      epack[i] = (uint32_t)(ecode[(e0 + i) * 3] | (ecode[(e0 + i) * 3 + 1] << 10)
                            | (ecode[(e0 + i) * 3 + 2] << 20));
    __syncthreads();
    if (r >= R) continue;
    for (int i = lane; i < SCALAR_ETILE && e0 + i < E; i += WAVE) {
      const uint32_t ec = epack[i];
      const uint32_t x = rc ^ ec;
      float s = 0.0f;
      if ((x & 0x3FFu) == 0) s += b0;
      if ((x & 0xFFC00u) == 0) s += b1;
      if ((x & 0x3FF00000u) == 0) s += b2;
      score[(int64_t)r * E + e0 + i] = s;
    }
  }
}

void mfma_score_bench(torch::Tensor rcode, torch::Tensor rbonus,
                      torch::Tensor ecode, int64_t K, torch::Tensor score) {
  const int R = (int)rcode.size(0), E = (int)ecode.size(0);
  const int tiles = (R / 16) * (E / 16);
  hipLaunchKernelGGL(mfma_score_kernel, dim3((tiles + 3) / 4), dim3(256), 0,
                     at::cuda::getCurrentCUDAStream(),
                     rcode.data_ptr<int32_t>(), rbonus.data_ptr<float>(),
                     ecode.data_ptr<int32_t>(), R, E, (int)K,
                     score.data_ptr<float>());
}

void scalar_score_bench(torch::Tensor rcode, torch::Tensor rbonus,
                        torch::Tensor ecode, torch::Tensor score) {
  const int R = (int)rcode.size(0), E = (int)ecode.size(0);
  hipLaunchKernelGGL(scalar_score_kernel, dim3((R + 3) / 4), dim3(256), 0,
                     at::cuda::getCurrentCUDAStream(),
                     rcode.data_ptr<int32_t>(), rbonus.data_ptr<float>(),
                     ecode.data_ptr<int32_t>(), R, E, score.data_ptr<float>());
}

// ---------------------------------------------------------------------------
// K9a: KD-tree descent (flat tree, partitioning.py as_flat layout)
// ---------------------------------------------------------------------------

__global__ void kd_descent_kernel(
    const int32_t* __restrict__ ent_values, const int32_t* __restrict__ node_kind,
    const int32_t* __restrict__ node_attr, const int32_t* __restrict__ node_a,
    const int32_t* __restrict__ node_b, const int32_t* __restrict__ rset,
    int64_t E, int A, int32_t* __restrict__ ent_part_out) {
  const int64_t e = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (e >= E) return;
  int nid = 0;
  while (node_kind[nid] != 0) {
    const int a = node_attr[nid];
    const int32_t v = ent_values[e * A + a];
    bool right;
    if (node_kind[nid] == 1) {
      right = v > node_a[nid];
    } else {
      const int lo = node_a[nid], n = node_b[nid];
      right = contains_i32(rset, lo, lo + n, v);
    }
    nid = 2 * nid + (right ? 2 : 1);
  }
  ent_part_out[e] = node_a[nid];
}

// ---------------------------------------------------------------------------
// K1 (GPU): banded Levenshtein sim-pair sweep over a value domain
// ---------------------------------------------------------------------------

// One WAVE per value i: the wave builds a Myers bit-parallel Peq table for
// string i in LDS once, then its 64 lanes sweep candidate values j from the
// length-bucketed window (sim > 0 bounds |len_i - len_j|). Edit distance via
// Hyyro's bit-vector recurrence: O(len_j) 64-bit ops per pair instead of the
// O(len_i * len_j) scratch-array DP — ~25x fewer operations, no scratch.
constexpr int SP_WAVES = 4;  // waves per block

__device__ int myers_distance(const unsigned long long* Peq, int m,
                              const uint8_t* b, int lb) {
  if (m == 0) return lb;
  int score = m;
  unsigned long long Pv = ~0ull, Mv = 0ull;
  const unsigned long long last = 1ull << (m - 1);
  for (int j = 0; j < lb; ++j) {
    const unsigned long long Eq = Peq[b[j]];
    const unsigned long long Xv = Eq | Mv;
    const unsigned long long Xh = (((Eq & Pv) + Pv) ^ Pv) | Eq;
    unsigned long long Ph = Mv | ~(Xh | Pv);
    unsigned long long Mh = Pv & Xh;
    if (Ph & last) ++score;
    if (Mh & last) --score;
    Ph = (Ph << 1) | 1ull;
    Mh <<= 1;
    Pv = Mh | ~(Xv | Ph);
    Mv = Ph & Xv;
  }
  return score;
}

__global__ void __launch_bounds__(SP_WAVES * WAVE) sim_pairs_kernel(
    const uint8_t* __restrict__ strs,  // [V, 64]
    const int32_t* __restrict__ lens,  // [V]
    const int32_t* __restrict__ len_order,  // [V] value ids sorted by length
    const int64_t* __restrict__ len_ptr,    // [66] bucket offsets by length
    int V, int max_len, float threshold, float max_sim,
    int64_t* __restrict__ row_counts,  // [V] (pass 1 out) or row offsets (pass 2 in)
    int32_t* __restrict__ out_col,     // pass 2
    float* __restrict__ out_expsim,    // pass 2 (stores exp(sim))
    int64_t* __restrict__ fill_pos,    // [V] atomic cursors (pass 2)
    int fill) {
  __shared__ unsigned long long peq_block[SP_WAVES][256];
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const int i = blockIdx.x * SP_WAVES + wave;
  if (i >= V) return;
  unsigned long long* Peq = peq_block[wave];
  const int la = lens[i];
  const uint8_t* a = strs + (int64_t)i * max_len;
  for (int c = lane; c < 256; c += WAVE) Peq[c] = 0ull;
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_wave_barrier();
  if (lane < la) atomicOr(&Peq[a[lane]], 1ull << lane);
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_wave_barrier();

  const float u0 = threshold / max_sim;
  const float scale = max_sim / (max_sim - threshold);
  const float band = (1.0f - u0) / (1.0f + u0);
  // length window: |la - lb| < band * (la + lb)
  int lb_min, lb_max;
  if (band >= 0.999f) {  // threshold ~ 0: no length pruning possible
    lb_min = 0;
    lb_max = max_len;
  } else {
    lb_min = (int)ceilf((1.0f - band) / (1.0f + band) * (float)la);
    lb_max = (int)floorf((1.0f + band) / (1.0f - band) * (float)la);
  }
  if (la == 0) { lb_min = 0; lb_max = 0; }
  if (lb_min < 0) lb_min = 0;
  if (lb_max > max_len) lb_max = max_len;
  const int64_t j0 = len_ptr[lb_min], j1 = len_ptr[lb_max + 1];

  int64_t n_hits = 0;
  for (int64_t jj = j0 + lane; jj < j1; jj += WAVE) {
    const int j = len_order[jj];
    const int lb = lens[j];
    const int tot = la + lb;
    float unit;
    if (tot == 0) {
      unit = 1.0f;
    } else {
      if ((float)abs(la - lb) >= band * (float)tot) continue;
      const int d = myers_distance(Peq, la, strs + (int64_t)j * max_len, lb);
      unit = 1.0f - 2.0f * (float)d / ((float)tot + (float)d);
    }
    const float trans = scale * (max_sim * unit - threshold);
    if (trans <= 0.0f) continue;  // expsim <= 1 filtered (AttributeIndex.scala:226)
    if (!fill) {
      ++n_hits;
    } else {
      const int64_t pos = atomicAdd((unsigned long long*)&fill_pos[i], 1ull) + row_counts[i];
      out_col[pos] = j;
      out_expsim[pos] = __expf(trans);
    }
  }
  if (!fill) {
    double total = wave_sum((double)n_hits);
    if (lane == 0) row_counts[i] = (int64_t)(total + 0.5);
  }
}

// ---------------------------------------------------------------------------
// host launchers
// ---------------------------------------------------------------------------

static int64_t wave_grid(int64_t items, int waves_per_block) {
  return (items + waves_per_block - 1) / waves_per_block;
}

// LDS sim-row staging in the link kernels; DBLINK_SIMH=0 restores the
// global-memory binary-search path (A/B; results are bitwise identical).
// Re-read per launch so one process can time both arms.
static int simh_enabled() {
  const char* e = std::getenv("DBLINK_SIMH");
  return e ? std::atoi(e) : 1;
}

void link_update(
    torch::Tensor rec_values, torch::Tensor rec_dist, torch::Tensor rec_gid,
    torch::Tensor rec_part, torch::Tensor cand_lo, torch::Tensor cand_hi,
    torch::Tensor postings, torch::Tensor ent_values, torch::Tensor ent_ptr,
    torch::Tensor log_norm, torch::Tensor voff, torch::Tensor csr_row_ptr,
    torch::Tensor csr_col, torch::Tensor csr_sim, torch::Tensor attr_const,
    int64_t seed, int64_t iteration, torch::Tensor rec_ent_out,
    torch::Tensor rec_ent_in, torch::Tensor error_count,
    torch::Tensor small_mask, torch::Tensor ctrl, torch::Tensor pair_a1,
    torch::Tensor pair_a2) {
  const int64_t* ctrl_ptr = ctrl.numel() ? ctrl.data_ptr<int64_t>() : nullptr;
  const int NP = (int)pair_a1.numel();
  CHECK_GPU(rec_values);
  const int64_t R = rec_values.size(0);
  const int A = (int)rec_values.size(1);
  TORCH_CHECK(A <= MAX_ATTRS, "at most ", MAX_ATTRS, " matching attributes supported");
  if (R == 0) return;
  constexpr int WPB = 4;
  const bool split = small_mask.numel() > 0;
  const uint8_t* mask_ptr = split ? small_mask.data_ptr<uint8_t>() : nullptr;
  {
    dim3 grid((unsigned)wave_grid(R, WPB));
    hipLaunchKernelGGL(link_update_kernel, grid, dim3(WPB * WAVE), 0,
                       at::cuda::getCurrentCUDAStream(),
                       rec_values.data_ptr<int32_t>(), rec_dist.data_ptr<uint8_t>(),
                       rec_gid.data_ptr<int64_t>(), rec_part.data_ptr<int32_t>(),
                       cand_lo.data_ptr<int64_t>(), cand_hi.data_ptr<int64_t>(),
                       postings.data_ptr<int32_t>(), ent_values.data_ptr<int32_t>(),
                       ent_ptr.data_ptr<int64_t>(), log_norm.data_ptr<float>(),
                       voff.data_ptr<int64_t>(), csr_row_ptr.data_ptr<int64_t>(),
                       csr_col.data_ptr<int32_t>(), csr_sim.data_ptr<float>(),
                       attr_const.data_ptr<uint8_t>(),
                       pair_a1.data_ptr<int32_t>(), pair_a2.data_ptr<int32_t>(),
                       NP, mask_ptr, R, A,
                       (uint64_t)seed, (uint32_t)iteration, ctrl_ptr,
                       rec_ent_out.data_ptr<int64_t>(),
                       rec_ent_in.data_ptr<int64_t>(), error_count.data_ptr<int>(),
                       simh_enabled());
  }
  if (split) {
    dim3 grid((unsigned)((R + 255) / 256));
    hipLaunchKernelGGL(link_update_small_kernel, grid, dim3(256), 0,
                       at::cuda::getCurrentCUDAStream(),
                       mask_ptr, R,
                       rec_values.data_ptr<int32_t>(), rec_dist.data_ptr<uint8_t>(),
                       rec_gid.data_ptr<int64_t>(), cand_lo.data_ptr<int64_t>(),
                       cand_hi.data_ptr<int64_t>(), postings.data_ptr<int32_t>(),
                       ent_values.data_ptr<int32_t>(), log_norm.data_ptr<float>(),
                       voff.data_ptr<int64_t>(), csr_row_ptr.data_ptr<int64_t>(),
                       csr_col.data_ptr<int32_t>(), csr_sim.data_ptr<float>(),
                       attr_const.data_ptr<uint8_t>(),
                       pair_a1.data_ptr<int32_t>(), pair_a2.data_ptr<int32_t>(),
                       NP, A, (uint64_t)seed,
                       (uint32_t)iteration, ctrl_ptr,
                       rec_ent_out.data_ptr<int64_t>(),
                       rec_ent_in.data_ptr<int64_t>(), error_count.data_ptr<int>());
  }
}

void postings_hist(torch::Tensor ent_part, torch::Tensor ent_values,
                   torch::Tensor pair_a1, torch::Tensor pair_a2,
                   torch::Tensor pair_v2, int64_t Vmax, torch::Tensor counts) {
  const int64_t E = ent_values.size(0);
  const int A = (int)ent_values.size(1);
  const int NP = (int)pair_a1.numel();
  const int64_t n = E * (A + NP);
  if (n == 0) return;
  TORCH_CHECK(counts.numel() <= INT32_MAX, "dense key space exceeds int32");
  dim3 grid((unsigned)((n + PB_THREADS - 1) / PB_THREADS));
  hipLaunchKernelGGL(postings_hist_kernel, grid, dim3(PB_THREADS), 0,
                     at::cuda::getCurrentCUDAStream(),
                     ent_part.data_ptr<int32_t>(), ent_values.data_ptr<int32_t>(),
                     pair_a1.data_ptr<int32_t>(), pair_a2.data_ptr<int32_t>(),
                     pair_v2.data_ptr<int32_t>(), E, A, NP, Vmax,
                     counts.data_ptr<int32_t>());
}

void postings_scatter(torch::Tensor ent_part, torch::Tensor ent_values,
                      torch::Tensor pair_a1, torch::Tensor pair_a2,
                      torch::Tensor pair_v2, int64_t Vmax, torch::Tensor cursor,
                      torch::Tensor postings) {
  const int64_t E = ent_values.size(0);
  const int A = (int)ent_values.size(1);
  const int NP = (int)pair_a1.numel();
  const int64_t n = E * (A + NP);
  if (n == 0) return;
  TORCH_CHECK(cursor.numel() <= INT32_MAX, "dense key space exceeds int32");
  dim3 grid((unsigned)((n + PB_THREADS - 1) / PB_THREADS));
  hipLaunchKernelGGL(postings_scatter_kernel, grid, dim3(PB_THREADS), 0,
                     at::cuda::getCurrentCUDAStream(),
                     ent_part.data_ptr<int32_t>(), ent_values.data_ptr<int32_t>(),
                     pair_a1.data_ptr<int32_t>(), pair_a2.data_ptr<int32_t>(),
                     pair_v2.data_ptr<int32_t>(), E, A, NP, Vmax,
                     cursor.data_ptr<int32_t>(), postings.data_ptr<int32_t>());
}

void classify_modes(torch::Tensor rec_values, torch::Tensor rec_dist,
                    torch::Tensor rec_part, torch::Tensor ent_ptr,
                    torch::Tensor cand_lo, torch::Tensor cand_hi, int64_t NP,
                    int64_t small_threshold, int64_t heavy_threshold,
                    torch::Tensor mode) {
  const int64_t R = rec_values.size(0);
  const int A = (int)rec_values.size(1);
  if (R == 0) return;
  dim3 grid((unsigned)((R + 255) / 256));
  hipLaunchKernelGGL(classify_modes_kernel, grid, dim3(256), 0,
                     at::cuda::getCurrentCUDAStream(),
                     rec_values.data_ptr<int32_t>(), rec_dist.data_ptr<uint8_t>(),
                     rec_part.data_ptr<int32_t>(), ent_ptr.data_ptr<int64_t>(),
                     cand_lo.data_ptr<int64_t>(), cand_hi.data_ptr<int64_t>(),
                     R, A, (int)NP, small_threshold, heavy_threshold,
                     mode.data_ptr<uint8_t>());
}

// Stable inverted-index keys: (key * E + e) sorts into the same segment
// boundaries as the counting-sort prefix, but with entities in ascending-id
// order within each segment — deterministic indexed draws for the heavy
// sampler's posting-segment proposals.
__global__ void build_ekeys_stable_kernel(
    const int32_t* __restrict__ ent_part, const int32_t* __restrict__ ent_values,
    const int32_t* __restrict__ pair_a1, const int32_t* __restrict__ pair_a2,
    const int32_t* __restrict__ pair_v2, int64_t E, int A, int NP, int64_t Vmax,
    int64_t* __restrict__ ekeys) {
  const int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int T = A + NP;
  if (idx >= E * T) return;
  const int t = (int)(idx / E);
  const int64_t e = idx % E;
  const int64_t v = posting_slot_value(ent_values, e, A, t, pair_a1, pair_a2, pair_v2);
  const int64_t key = ((int64_t)ent_part[e] * T + t) * Vmax + v;
  ekeys[idx] = key * E + e;
}

// hipGraph-safe radix sort: torch.sort allocates its workspace through the
// caching allocator, which faults on replay when capture records multi-GB
// pools (the r01 ">4M records" graph bug). These wrappers let the engine own
// ONE persistent workspace and keep every launch on the capture stream.
int64_t radix_sort_pairs_temp_bytes(int64_t n) {
  size_t bytes = 0;
  rocprim::radix_sort_pairs(
      nullptr, bytes, (const int64_t*)nullptr, (int64_t*)nullptr,
      (const int32_t*)nullptr, (int32_t*)nullptr, (size_t)n);
  return (int64_t)bytes;
}

void radix_sort_pairs_i64_i32(torch::Tensor keys_in, torch::Tensor keys_out,
                              torch::Tensor vals_in, torch::Tensor vals_out,
                              int64_t end_bit, torch::Tensor temp) {
  const size_t n = (size_t)keys_in.numel();
  if (n == 0) return;
  size_t bytes = (size_t)temp.numel();
  hipError_t err = rocprim::radix_sort_pairs(
      temp.data_ptr(), bytes, keys_in.data_ptr<int64_t>(),
      keys_out.data_ptr<int64_t>(), vals_in.data_ptr<int32_t>(),
      vals_out.data_ptr<int32_t>(), n, 0, (unsigned)end_bit,
      at::cuda::getCurrentCUDAStream());
  TORCH_CHECK(err == hipSuccess, "radix_sort_pairs failed: ",
              hipGetErrorString(err));
}

void build_ekeys_stable(torch::Tensor ent_part, torch::Tensor ent_values,
                        torch::Tensor pair_a1, torch::Tensor pair_a2,
                        torch::Tensor pair_v2, int64_t Vmax,
                        torch::Tensor ekeys) {
  const int64_t E = ent_values.size(0);
  const int A = (int)ent_values.size(1);
  const int NP = (int)pair_a1.numel();
  const int64_t n = E * (A + NP);
  if (n == 0) return;
  dim3 grid((unsigned)((n + 255) / 256));
  hipLaunchKernelGGL(build_ekeys_stable_kernel, grid, dim3(256), 0,
                     at::cuda::getCurrentCUDAStream(),
                     ent_part.data_ptr<int32_t>(), ent_values.data_ptr<int32_t>(),
                     pair_a1.data_ptr<int32_t>(), pair_a2.data_ptr<int32_t>(),
                     pair_v2.data_ptr<int32_t>(), E, A, NP, Vmax,
                     ekeys.data_ptr<int64_t>());
}

void link_update_heavy(
    torch::Tensor mode, torch::Tensor rec_values, torch::Tensor rec_dist,
    torch::Tensor rec_gid, torch::Tensor rec_part, torch::Tensor ent_values,
    torch::Tensor ent_ptr, torch::Tensor log_norm, torch::Tensor voff,
    torch::Tensor csr_row_ptr, torch::Tensor csr_col, torch::Tensor csr_sim,
    torch::Tensor attr_const, torch::Tensor csr_row_ptr_big,
    torch::Tensor csr_col_big, torch::Tensor csr_sim_big, double tau,
    torch::Tensor postings, torch::Tensor idx_ptr,
    int64_t Vmax, int64_t NP, int64_t seed, int64_t iteration, torch::Tensor ctrl,
    torch::Tensor rec_ent_out, torch::Tensor rec_ent_in,
    torch::Tensor error_count, torch::Tensor stats) {
  const int64_t* ctrl_ptr = ctrl.numel() ? ctrl.data_ptr<int64_t>() : nullptr;
  const int64_t R = rec_values.size(0);
  const int A = (int)rec_values.size(1);
  if (R == 0) return;
  dim3 grid((unsigned)wave_grid(R, HEAVY_WAVES));
  hipLaunchKernelGGL(link_update_heavy_kernel, grid, dim3(HEAVY_WAVES * WAVE), 0,
                     at::cuda::getCurrentCUDAStream(),
                     mode.data_ptr<uint8_t>(), rec_values.data_ptr<int32_t>(),
                     rec_dist.data_ptr<uint8_t>(), rec_gid.data_ptr<int64_t>(),
                     rec_part.data_ptr<int32_t>(), ent_values.data_ptr<int32_t>(),
                     ent_ptr.data_ptr<int64_t>(), log_norm.data_ptr<float>(),
                     voff.data_ptr<int64_t>(), csr_row_ptr.data_ptr<int64_t>(),
                     csr_col.data_ptr<int32_t>(), csr_sim.data_ptr<float>(),
                     attr_const.data_ptr<uint8_t>(),
                     csr_row_ptr_big.data_ptr<int64_t>(),
                     csr_col_big.data_ptr<int32_t>(), csr_sim_big.data_ptr<float>(),
                     (float)tau, postings.data_ptr<int32_t>(),
                     idx_ptr.data_ptr<int64_t>(), Vmax, (int)NP, R, A,
                     (uint64_t)seed, (uint32_t)iteration, ctrl_ptr,
                     rec_ent_out.data_ptr<int64_t>(), rec_ent_in.data_ptr<int64_t>(),
                     error_count.data_ptr<int>(),
                     stats.numel() ? (unsigned long long*)stats.data_ptr<int64_t>()
                                   : nullptr,
                     simh_enabled());
}

void cand_ranges(torch::Tensor rec_part, torch::Tensor rec_values,
                 torch::Tensor pair_a1, torch::Tensor pair_a2,
                 torch::Tensor pair_v2, torch::Tensor ptr, int64_t Vmax,
                 torch::Tensor cand_lo, torch::Tensor cand_hi) {
  const int64_t R = rec_values.size(0);
  const int A = (int)rec_values.size(1);
  const int NP = (int)pair_a1.numel();
  const int64_t n = R * (A + NP);
  if (n == 0) return;
  dim3 grid((unsigned)((n + 255) / 256));
  hipLaunchKernelGGL(cand_ranges_kernel, grid, dim3(256), 0,
                     at::cuda::getCurrentCUDAStream(),
                     rec_part.data_ptr<int32_t>(), rec_values.data_ptr<int32_t>(),
                     pair_a1.data_ptr<int32_t>(), pair_a2.data_ptr<int32_t>(),
                     pair_v2.data_ptr<int32_t>(), ptr.data_ptr<int64_t>(),
                     R, A, NP, Vmax,
                     cand_lo.data_ptr<int64_t>(), cand_hi.data_ptr<int64_t>());
}

void link_update_dense(
    torch::Tensor rec_values, torch::Tensor rec_dist, torch::Tensor rec_gid,
    torch::Tensor rec_part, torch::Tensor rec_file, torch::Tensor ent_values,
    torch::Tensor ent_ptr, torch::Tensor theta, torch::Tensor phi,
    torch::Tensor norm_lin, torch::Tensor voff, torch::Tensor csr_row_ptr,
    torch::Tensor csr_col, torch::Tensor csr_sim, torch::Tensor attr_const,
    int64_t collapsed, int64_t seed, int64_t iteration, torch::Tensor rec_ent_out,
    torch::Tensor ctrl) {
  const int64_t* ctrl_ptr = ctrl.numel() ? ctrl.data_ptr<int64_t>() : nullptr;
  const int64_t R = rec_values.size(0);
  const int A = (int)rec_values.size(1);
  const int F = (int)theta.size(1);
  if (R == 0) return;
  constexpr int WPB = 4;
  dim3 grid((unsigned)wave_grid(R, WPB));
  hipLaunchKernelGGL(link_update_dense_kernel, grid, dim3(WPB * WAVE), 0,
                     at::cuda::getCurrentCUDAStream(),
                     rec_values.data_ptr<int32_t>(), rec_dist.data_ptr<uint8_t>(),
                     rec_gid.data_ptr<int64_t>(), rec_part.data_ptr<int32_t>(),
                     rec_file.data_ptr<int32_t>(), ent_values.data_ptr<int32_t>(),
                     ent_ptr.data_ptr<int64_t>(), theta.data_ptr<float>(),
                     phi.data_ptr<float>(), norm_lin.data_ptr<float>(),
                     voff.data_ptr<int64_t>(), csr_row_ptr.data_ptr<int64_t>(),
                     csr_col.data_ptr<int32_t>(), csr_sim.data_ptr<float>(),
                     attr_const.data_ptr<uint8_t>(), R, A, F, (int)collapsed,
                     (uint64_t)seed, (uint32_t)iteration, ctrl_ptr,
                     rec_ent_out.data_ptr<int64_t>(), simh_enabled());
}

// Opt-in value-phase work counters (DBLink_VALUE_STATS): set once from
// python; the tensor is held so the device pointer stays alive.
static unsigned long long* g_value_stats = nullptr;
static torch::Tensor g_value_stats_keep;
static const double* g_tab_excl = nullptr;
static const double* g_tab_rawsum = nullptr;
static const float* g_self_expsim = nullptr;
static int g_ktab_max = 0;
static int64_t g_tab_nnz = 0;
static std::vector<torch::Tensor> g_ktab_keep;
static const double* g_tab2_excl = nullptr;
static const double* g_tab2_rawsum = nullptr;
static int g_k2tab_max = 0;
static std::vector<torch::Tensor> g_k2tab_keep;

void set_value_k2tables(torch::Tensor excl, torch::Tensor rawsum, int64_t k2max) {
  if (k2max >= 2 && excl.numel()) {
    g_k2tab_keep = {excl, rawsum};
    g_tab2_excl = excl.data_ptr<double>();
    g_tab2_rawsum = rawsum.data_ptr<double>();
    g_k2tab_max = (int)k2max;
  } else {
    g_k2tab_keep.clear();
    g_tab2_excl = nullptr;
    g_tab2_rawsum = nullptr;
    g_k2tab_max = 0;
  }
}

// Install the k>=2 single-distinct-value perturbation tables (one level per
// k in [2, kmax]); pairs whose linked records all share one (value, file)
// then draw in O(log row) like the k=1 kernel.
void set_value_ktables(torch::Tensor excl, torch::Tensor rawsum,
                       torch::Tensor self_expsim, int64_t kmax, int64_t nnz) {
  if (kmax >= 2 && excl.numel()) {
    g_ktab_keep = {excl, rawsum, self_expsim};
    g_tab_excl = excl.data_ptr<double>();
    g_tab_rawsum = rawsum.data_ptr<double>();
    g_self_expsim = self_expsim.data_ptr<float>();
    g_ktab_max = (int)kmax;
    g_tab_nnz = nnz;
  } else {
    g_ktab_keep.clear();
    g_tab_excl = nullptr;
    g_tab_rawsum = nullptr;
    g_self_expsim = nullptr;
    g_ktab_max = 0;
    g_tab_nnz = 0;
  }
}

void set_value_stats(torch::Tensor t) {
  if (t.numel()) {
    g_value_stats_keep = t;
    g_value_stats = (unsigned long long*)t.data_ptr<int64_t>();
  } else {
    g_value_stats = nullptr;
    g_value_stats_keep = torch::Tensor();
  }
}

static ValueArgs make_value_args(
    torch::Tensor rec_values, torch::Tensor rec_dist, torch::Tensor rec_file,
    torch::Tensor ent_rec_ptr, torch::Tensor ent_rec_idx, torch::Tensor ent_values,
    torch::Tensor theta, torch::Tensor phi, torch::Tensor log_phi,
    torch::Tensor norm_lin, torch::Tensor log_norm, torch::Tensor voff,
    torch::Tensor csr_row_ptr, torch::Tensor csr_col, torch::Tensor csr_sim,
    torch::Tensor phi_prob, torch::Tensor phi_alias, torch::Tensor pow_prob,
    torch::Tensor pow_alias, torch::Tensor pow_off, torch::Tensor log_pow_total,
    torch::Tensor attr_const, int64_t Kc, int64_t collapsed, int64_t seed,
    int64_t iteration, int64_t ent_id_base, torch::Tensor error_count,
    torch::Tensor csr_excl, torch::Tensor csr_rawsum, torch::Tensor z1) {
  ValueArgs a;
  a.ctrl = nullptr;
  a.kobs = nullptr;
  a.pair_list = nullptr;
  a.stats = g_value_stats;
  {
    const char* e = std::getenv("DBLINK_VCHUNK");
    a.vchunk = e ? std::atoi(e) : 1;
  }
  a.tab_excl = g_tab_excl;
  a.tab_rawsum = g_tab_rawsum;
  a.self_expsim = g_self_expsim;
  a.ktab_max = g_ktab_max;
  a.nnz = g_tab_nnz;
  a.tab2_excl = g_tab2_excl;
  a.tab2_rawsum = g_tab2_rawsum;
  a.k2tab_max = g_k2tab_max;
  a.csr_excl = csr_excl.numel() ? csr_excl.data_ptr<double>() : nullptr;
  a.csr_rawsum = csr_rawsum.data_ptr<double>();
  a.z1 = z1.data_ptr<double>();
  a.rec_values = rec_values.data_ptr<int32_t>();
  a.rec_dist = rec_dist.data_ptr<uint8_t>();
  a.rec_file = rec_file.data_ptr<int32_t>();
  a.ent_rec_ptr = ent_rec_ptr.data_ptr<int64_t>();
  a.ent_rec_idx = ent_rec_idx.data_ptr<int64_t>();
  a.ent_values = ent_values.data_ptr<int32_t>();
  a.theta = theta.data_ptr<float>();
  a.phi = phi.data_ptr<float>();
  a.log_phi = log_phi.data_ptr<float>();
  a.norm_lin = norm_lin.data_ptr<float>();
  a.log_norm = log_norm.data_ptr<float>();
  a.voff = voff.data_ptr<int64_t>();
  a.csr_row_ptr = csr_row_ptr.data_ptr<int64_t>();
  a.csr_col = csr_col.data_ptr<int32_t>();
  a.csr_sim = csr_sim.data_ptr<float>();
  a.phi_prob = phi_prob.data_ptr<float>();
  a.phi_alias = phi_alias.data_ptr<int32_t>();
  a.pow_prob = pow_prob.numel() ? pow_prob.data_ptr<float>() : nullptr;
  a.pow_alias = pow_alias.numel() ? pow_alias.data_ptr<int32_t>() : nullptr;
  a.pow_off = pow_off.data_ptr<int64_t>();
  a.log_pow_total = log_pow_total.data_ptr<float>();
  a.attr_const = attr_const.data_ptr<uint8_t>();
  a.Kc = (int)Kc;
  a.E = ent_values.size(0);
  a.A = (int)ent_values.size(1);
  a.F = (int)theta.size(1);
  a.collapsed = (int)collapsed;
  a.seed = (uint64_t)seed;
  a.iteration = (uint32_t)iteration;
  a.ent_id_base = (uint64_t)ent_id_base;
  a.error_count = error_count.data_ptr<int>();
  return a;
}

void value_update(
    torch::Tensor rec_values, torch::Tensor rec_dist, torch::Tensor rec_file,
    torch::Tensor ent_rec_ptr, torch::Tensor ent_rec_idx, torch::Tensor ent_values,
    torch::Tensor theta, torch::Tensor phi, torch::Tensor log_phi,
    torch::Tensor norm_lin, torch::Tensor log_norm, torch::Tensor voff,
    torch::Tensor csr_row_ptr, torch::Tensor csr_col, torch::Tensor csr_sim,
    torch::Tensor phi_prob, torch::Tensor phi_alias, torch::Tensor pow_prob,
    torch::Tensor pow_alias, torch::Tensor pow_off, torch::Tensor log_pow_total,
    torch::Tensor attr_const, int64_t Kc, int64_t collapsed, int64_t sequential,
    int64_t seed, int64_t iteration, int64_t ent_id_base, torch::Tensor error_count,
    torch::Tensor wave_pairs, torch::Tensor base_pairs, torch::Tensor k1_pairs,
    torch::Tensor csr_excl, torch::Tensor csr_rawsum, torch::Tensor z1,
    torch::Tensor ctrl, torch::Tensor kobs) {
  ValueArgs args = make_value_args(
      rec_values, rec_dist, rec_file, ent_rec_ptr, ent_rec_idx, ent_values, theta,
      phi, log_phi, norm_lin, log_norm, voff, csr_row_ptr, csr_col, csr_sim,
      phi_prob, phi_alias, pow_prob, pow_alias, pow_off, log_pow_total, attr_const,
      Kc, collapsed, seed, iteration, ent_id_base, error_count, csr_excl,
      csr_rawsum, z1);
  args.ctrl = ctrl.numel() ? ctrl.data_ptr<int64_t>() : nullptr;
  if (sequential) {
    const int64_t pairs = args.E * args.A;
    if (pairs == 0) return;
    args.pair_list = nullptr;
    args.n_pairs = pairs;
    dim3 grid((unsigned)wave_grid(pairs, 4));
    hipLaunchKernelGGL(value_update_seq_kernel, grid, dim3(4 * WAVE), 0,
                       at::cuda::getCurrentCUDAStream(), args);
    return;
  }
  if (kobs.numel() > 0) {
    // kobs self-selection: no host-side pair lists, no stream sync — the
    // four kernels each cover every pair and act only on their class
    // (k=0 base, k=1, k>=2 hash, k>=2 union-merge)
    args.kobs = kobs.data_ptr<int32_t>();
    args.n_pairs = kobs.numel();
    dim3 tgrid((unsigned)((args.n_pairs + 255) / 256));
    hipLaunchKernelGGL(value_base_draw_kernel, tgrid, dim3(256), 0,
                       at::cuda::getCurrentCUDAStream(), args);
    hipLaunchKernelGGL(value_update_k1_kernel, tgrid, dim3(256), 0,
                       at::cuda::getCurrentCUDAStream(), args);
    if (args.tab_excl != nullptr)
      hipLaunchKernelGGL(value_update_kd1_kernel, tgrid, dim3(256), 0,
                         at::cuda::getCurrentCUDAStream(), args);
    if (args.tab2_excl != nullptr) {
      const int64_t nw2 = (args.n_pairs + 7) / 8;
      hipLaunchKernelGGL(value_update_kd2_kernel,
                         dim3((unsigned)wave_grid(nw2, WAVES_PER_BLOCK_VAL)),
                         dim3(WAVES_PER_BLOCK_VAL * WAVE), 0,
                         at::cuda::getCurrentCUDAStream(), args);
    }
    const int64_t n_waves = (args.n_pairs + VAL_STRIDE - 1) / VAL_STRIDE;
    dim3 wgrid((unsigned)wave_grid(n_waves, WAVES_PER_BLOCK_VAL));
    hipLaunchKernelGGL(value_update_kernel_t<1>, wgrid,
                       dim3(WAVES_PER_BLOCK_VAL * WAVE), 0,
                       at::cuda::getCurrentCUDAStream(), args);
    hipLaunchKernelGGL(value_update_kernel_t<2>, wgrid,
                       dim3(WAVES_PER_BLOCK_VAL * WAVE), 0,
                       at::cuda::getCurrentCUDAStream(), args);
    return;
  }
  // explicit pair lists (tests / special cases)
  if (base_pairs.numel() > 0) {
    args.pair_list = base_pairs.data_ptr<int64_t>();
    args.n_pairs = base_pairs.numel();
    dim3 grid((unsigned)((args.n_pairs + 255) / 256));
    hipLaunchKernelGGL(value_base_draw_kernel, grid, dim3(256), 0,
                       at::cuda::getCurrentCUDAStream(), args);
  }
  if (k1_pairs.numel() > 0) {
    args.pair_list = k1_pairs.data_ptr<int64_t>();
    args.n_pairs = k1_pairs.numel();
    dim3 grid((unsigned)((args.n_pairs + 255) / 256));
    hipLaunchKernelGGL(value_update_k1_kernel, grid, dim3(256), 0,
                       at::cuda::getCurrentCUDAStream(), args);
  }
  if (wave_pairs.numel() > 0) {
    args.pair_list = wave_pairs.data_ptr<int64_t>();
    args.n_pairs = wave_pairs.numel();
    dim3 grid((unsigned)wave_grid(args.n_pairs, WAVES_PER_BLOCK_VAL));
    hipLaunchKernelGGL(value_update_kernel_t<0>, grid, dim3(WAVES_PER_BLOCK_VAL * WAVE), 0,
                       at::cuda::getCurrentCUDAStream(), args);
  }
}

void distortion_update(
    torch::Tensor rec_values, torch::Tensor rec_dist, torch::Tensor rec_file,
    torch::Tensor rec_gid, torch::Tensor rec_ent, torch::Tensor ent_values,
    torch::Tensor theta, torch::Tensor phi, torch::Tensor norm_lin,
    torch::Tensor self_expsim, torch::Tensor voff, torch::Tensor attr_const,
    int64_t seed, int64_t iteration, torch::Tensor ctrl, torch::Tensor log_phi,
    torch::Tensor log_norm, torch::Tensor csr_row_ptr, torch::Tensor csr_col,
    torch::Tensor csr_sim, torch::Tensor loglik) {
  const int64_t* ctrl_ptr = ctrl.numel() ? ctrl.data_ptr<int64_t>() : nullptr;
  const bool fuse = loglik.numel() > 0;
  TORCH_CHECK(!fuse || loglik.numel() >= 256, "loglik buffer must have 256 slots");
  const int64_t R = rec_values.size(0);
  const int A = (int)rec_values.size(1);
  const int F = (int)theta.size(1);
  const int64_t E = fuse ? ent_values.size(0) : 0;
  if (R == 0) return;
  const int64_t total = (E + R) * A;
  dim3 grid((unsigned)((total + 255) / 256));
  hipLaunchKernelGGL(distortion_update_kernel, grid, dim3(256), 0,
                     at::cuda::getCurrentCUDAStream(),
                     rec_values.data_ptr<int32_t>(), rec_dist.data_ptr<uint8_t>(),
                     rec_file.data_ptr<int32_t>(), rec_gid.data_ptr<int64_t>(),
                     rec_ent.data_ptr<int64_t>(), ent_values.data_ptr<int32_t>(),
                     theta.data_ptr<float>(), phi.data_ptr<float>(),
                     norm_lin.data_ptr<float>(), self_expsim.data_ptr<float>(),
                     voff.data_ptr<int64_t>(), attr_const.data_ptr<uint8_t>(),
                     R, A, F, (uint64_t)seed, (uint32_t)iteration, ctrl_ptr,
                     log_phi.data_ptr<float>(), log_norm.data_ptr<float>(),
                     csr_row_ptr.data_ptr<int64_t>(), csr_col.data_ptr<int32_t>(),
                     csr_sim.data_ptr<float>(), E,
                     fuse ? loglik.data_ptr<double>() : nullptr);
}

void summary_loglik(
    torch::Tensor ent_values, torch::Tensor rec_values, torch::Tensor rec_dist,
    torch::Tensor rec_ent, torch::Tensor log_phi, torch::Tensor log_norm,
    torch::Tensor voff, torch::Tensor csr_row_ptr, torch::Tensor csr_col,
    torch::Tensor csr_sim, torch::Tensor attr_const, torch::Tensor out) {
  TORCH_CHECK(out.numel() >= 256, "loglik buffer must have 256 slots");
  const int64_t E = ent_values.size(0);
  const int64_t R = rec_values.size(0);
  const int A = (int)ent_values.size(1);
  const int64_t total = (E + R) * A;
  if (total == 0) return;
  dim3 grid((unsigned)((total + 255) / 256));
  hipLaunchKernelGGL(summary_loglik_kernel, grid, dim3(256), 0,
                     at::cuda::getCurrentCUDAStream(),
                     ent_values.data_ptr<int32_t>(), rec_values.data_ptr<int32_t>(),
                     rec_dist.data_ptr<uint8_t>(), rec_ent.data_ptr<int64_t>(),
                     log_phi.data_ptr<float>(), log_norm.data_ptr<float>(),
                     voff.data_ptr<int64_t>(), csr_row_ptr.data_ptr<int64_t>(),
                     csr_col.data_ptr<int32_t>(), csr_sim.data_ptr<float>(),
                     attr_const.data_ptr<uint8_t>(), E, R, A,
                     out.data_ptr<double>());
}

void build_keys(torch::Tensor ent_part, torch::Tensor ent_values,
                torch::Tensor rec_part, torch::Tensor rec_values,
                torch::Tensor pair_a1, torch::Tensor pair_a2,
                torch::Tensor pair_v2, int64_t Vmax,
                torch::Tensor ekeys, torch::Tensor qkeys) {
  const int64_t E = ent_values.size(0);
  const int64_t R = rec_values.size(0);
  const int A = (int)ent_values.size(1);
  const int NP = (int)pair_a1.numel();
  const int64_t total = (E + R) * (A + NP);
  if (total == 0) return;
  dim3 grid((unsigned)((total + 255) / 256));
  hipLaunchKernelGGL(build_keys_kernel, grid, dim3(256), 0,
                     at::cuda::getCurrentCUDAStream(),
                     ent_part.data_ptr<int32_t>(), ent_values.data_ptr<int32_t>(),
                     rec_part.data_ptr<int32_t>(), rec_values.data_ptr<int32_t>(),
                     pair_a1.data_ptr<int32_t>(), pair_a2.data_ptr<int32_t>(),
                     pair_v2.data_ptr<int32_t>(),
                     E, R, A, NP, Vmax, ekeys.data_ptr<int64_t>(),
                     qkeys.data_ptr<int64_t>());
}

void summary_counts(torch::Tensor rec_dist, torch::Tensor rec_file,
                    torch::Tensor ent_rec_ptr, int64_t E, torch::Tensor counts,
                    torch::Tensor loglik, torch::Tensor packed) {
  const int64_t R = rec_dist.size(0);
  const int A = (int)rec_dist.size(1);
  const int F = (int)((counts.numel() - 1 - (A + 1)) / A);
  const int64_t total = E + R;
  const int n_counts = (int)counts.numel();
  if (total > 0) {
    dim3 grid((unsigned)((total + 255) / 256));
    hipLaunchKernelGGL(summary_counts_kernel, grid, dim3(256),
                       n_counts * sizeof(unsigned int),
                       at::cuda::getCurrentCUDAStream(),
                       rec_dist.data_ptr<uint8_t>(), rec_file.data_ptr<int32_t>(),
                       ent_rec_ptr.data_ptr<int64_t>(), E, R, A, F, n_counts,
                       (unsigned long long*)counts.data_ptr<int64_t>());
  }
  dim3 g2((unsigned)((n_counts + 255) / 256));
  hipLaunchKernelGGL(summary_finalize_kernel, g2, dim3(256), 0,
                     at::cuda::getCurrentCUDAStream(),
                     loglik.data_ptr<double>(),
                     (const unsigned long long*)counts.data_ptr<int64_t>(),
                     n_counts, packed.data_ptr<double>());
}

void kd_descent(
    torch::Tensor ent_values, torch::Tensor node_kind, torch::Tensor node_attr,
    torch::Tensor node_a, torch::Tensor node_b, torch::Tensor rset,
    torch::Tensor ent_part_out) {
  const int64_t E = ent_values.size(0);
  const int A = (int)ent_values.size(1);
  if (E == 0) return;
  dim3 grid((unsigned)((E + 255) / 256));
  hipLaunchKernelGGL(kd_descent_kernel, grid, dim3(256), 0,
                     at::cuda::getCurrentCUDAStream(),
                     ent_values.data_ptr<int32_t>(), node_kind.data_ptr<int32_t>(),
                     node_attr.data_ptr<int32_t>(), node_a.data_ptr<int32_t>(),
                     node_b.data_ptr<int32_t>(), rset.data_ptr<int32_t>(), E, A,
                     ent_part_out.data_ptr<int32_t>());
}

std::vector<torch::Tensor> sim_pairs_gpu(torch::Tensor strs, torch::Tensor lens,
                                         double threshold, double max_sim) {
  CHECK_GPU(strs);
  const int V = (int)strs.size(0);
  const int max_len = (int)strs.size(1);
  TORCH_CHECK(max_len <= 64, "attribute values longer than 64 bytes unsupported");
  auto opts_i64 = torch::TensorOptions().dtype(torch::kInt64).device(strs.device());
  // length buckets: candidates for value i live in a contiguous window of
  // the length-sorted order
  auto lens64 = lens.to(torch::kInt64);
  auto len_order = torch::argsort(lens64, /*stable=*/true).to(torch::kInt32).contiguous();
  auto sorted_lens = std::get<0>(torch::sort(lens64));
  auto len_ptr = torch::searchsorted(
      sorted_lens, torch::arange(66, opts_i64), /*out_int32=*/false, /*right=*/false)
      .contiguous();
  auto row_counts = torch::zeros({V}, opts_i64);
  dim3 grid((unsigned)((V + SP_WAVES - 1) / SP_WAVES));
  hipLaunchKernelGGL(sim_pairs_kernel, grid, dim3(SP_WAVES * WAVE), 0,
                     at::cuda::getCurrentCUDAStream(),
                     strs.data_ptr<uint8_t>(), lens.data_ptr<int32_t>(),
                     len_order.data_ptr<int32_t>(), len_ptr.data_ptr<int64_t>(),
                     V, max_len, (float)threshold, (float)max_sim,
                     row_counts.data_ptr<int64_t>(), nullptr, nullptr, nullptr, 0);
  auto row_ptr = torch::zeros({V + 1}, opts_i64);
  row_ptr.slice(0, 1, V + 1) = torch::cumsum(row_counts, 0);
  auto row_start = row_ptr.slice(0, 0, V).contiguous();
  const int64_t nnz = row_ptr[V].item<int64_t>();
  auto col = torch::empty({nnz}, torch::TensorOptions().dtype(torch::kInt32).device(strs.device()));
  auto expsim = torch::empty({nnz}, torch::TensorOptions().dtype(torch::kFloat32).device(strs.device()));
  auto fill_pos = torch::zeros({V}, opts_i64);
  hipLaunchKernelGGL(sim_pairs_kernel, grid, dim3(SP_WAVES * WAVE), 0,
                     at::cuda::getCurrentCUDAStream(),
                     strs.data_ptr<uint8_t>(), lens.data_ptr<int32_t>(),
                     len_order.data_ptr<int32_t>(), len_ptr.data_ptr<int64_t>(),
                     V, max_len, (float)threshold, (float)max_sim,
                     row_start.data_ptr<int64_t>(), col.data_ptr<int32_t>(),
                     expsim.data_ptr<float>(), fill_pos.data_ptr<int64_t>(), 1);
  return {row_ptr, col, expsim};
}

}  // namespace dblink
