// Common device/host utilities for dblink_amd kernels (gfx950 / CDNA4).
//
// Philox4x32-10 counter-based RNG: every random draw in the GPU sweep is a
// pure function of (seed, iteration, phase, element id, draw index), giving
// order-independent determinism — the MI355X replacement for the reference's
// per-partition MersenneTwister streams (GibbsUpdates.scala:139-147).
#pragma once

#include <cstdint>

#ifdef __HIPCC__
#define DBL_HD __host__ __device__ __forceinline__
#define DBL_D __device__ __forceinline__
#else
#define DBL_HD inline
#define DBL_D inline
#endif

namespace dblink {

struct u32x4 {
  uint32_t x, y, z, w;
};

DBL_HD uint32_t mulhilo(uint32_t a, uint32_t b, uint32_t* hi) {
  uint64_t p = (uint64_t)a * (uint64_t)b;
  *hi = (uint32_t)(p >> 32);
  return (uint32_t)p;
}

// Philox4x32-10 (Salmon et al. 2011, public-domain algorithm).
DBL_HD u32x4 philox4x32(uint64_t seed, uint32_t c0, uint32_t c1, uint32_t c2, uint32_t c3) {
  const uint32_t M0 = 0xD2511F53u, M1 = 0xCD9E8D57u;
  const uint32_t W0 = 0x9E3779B9u, W1 = 0xBB67AE85u;
  uint32_t k0 = (uint32_t)seed, k1 = (uint32_t)(seed >> 32);
  uint32_t x0 = c0, x1 = c1, x2 = c2, x3 = c3;
#pragma unroll
  for (int round = 0; round < 10; ++round) {
    uint32_t hi0, hi1;
    uint32_t lo0 = mulhilo(M0, x0, &hi0);
    uint32_t lo1 = mulhilo(M1, x2, &hi1);
    uint32_t y0 = hi1 ^ x1 ^ k0;
    uint32_t y1 = lo1;
    uint32_t y2 = hi0 ^ x3 ^ k1;
    uint32_t y3 = lo0;
    x0 = y0; x1 = y1; x2 = y2; x3 = y3;
    k0 += W0; k1 += W1;
  }
  return {x0, x1, x2, x3};
}

// Uniform in (0,1): never exactly 0 or 1 (safe for log()).
DBL_HD float u32_to_uniform(uint32_t v) {
  return ((float)v + 0.5f) * 2.3283064365386963e-10f;  // * 2^-32
}

// One uniform keyed by (seed | iter, phase | elem, draw).
DBL_HD float philox_uniform(uint64_t seed, uint32_t iter, uint32_t phase,
                            uint64_t elem, uint32_t draw) {
  u32x4 r = philox4x32(seed, (uint32_t)elem, (uint32_t)(elem >> 32),
                       iter ^ (phase << 24), draw);
  return u32_to_uniform(r.x);
}

// Two independent uniforms from one philox call.
DBL_HD void philox_uniform2(uint64_t seed, uint32_t iter, uint32_t phase,
                            uint64_t elem, uint32_t draw, float* u1, float* u2) {
  u32x4 r = philox4x32(seed, (uint32_t)elem, (uint32_t)(elem >> 32),
                       iter ^ (phase << 24), draw);
  *u1 = u32_to_uniform(r.x);
  *u2 = u32_to_uniform(r.y);
}

#ifdef __HIPCC__

constexpr int WAVE = 64;

DBL_D float gumbel_from_uniform(float u) { return -__logf(-__logf(u)); }

// Wave-wide argmax over (score, payload): returns the payload of the max
// score to every lane (ties broken toward lower lane).
DBL_D void wave_argmax(float& score, long long& payload) {
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) {
    float other_s = __shfl_down(score, off);
    long long other_p = __shfl_down(payload, off);
    if (other_s > score) {
      score = other_s;
      payload = other_p;
    }
  }
  score = __shfl(score, 0);
  payload = __shfl(payload, 0);
}

DBL_D double wave_sum(double v) {
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) v += __shfl_down(v, off);
  return __shfl(v, 0);
}

DBL_D float wave_sum_f32(float v) {
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) v += __shfl_down(v, off);
  return __shfl(v, 0);
}

// Binary search: first index in [lo, hi) with arr[i] >= key.
DBL_D int64_t lower_bound_i32(const int32_t* __restrict__ arr, int64_t lo, int64_t hi,
                              int32_t key) {
  while (lo < hi) {
    int64_t mid = (lo + hi) >> 1;
    if (arr[mid] < key) lo = mid + 1; else hi = mid;
  }
  return lo;
}

DBL_D bool contains_i32(const int32_t* __restrict__ arr, int64_t lo, int64_t hi,
                        int32_t key) {
  int64_t i = lower_bound_i32(arr, lo, hi, key);
  return i < hi && arr[i] == key;
}

// Sparse similarity lookup: sim(a: attr-row of x, col y) in LOG space
// (csr_sim stores the truncated similarity itself = log expsim); absent -> 0.
DBL_D float sim_lookup(const int64_t* __restrict__ row_ptr, const int32_t* __restrict__ col,
                       const float* __restrict__ sim, int64_t row, int32_t y) {
  int64_t lo = row_ptr[row], hi = row_ptr[row + 1];
  int64_t i = lower_bound_i32(col, lo, hi, y);
  return (i < hi && col[i] == y) ? sim[i] : 0.0f;
}

// Alias-table draw with two uniforms.
DBL_D int alias_draw(const float* __restrict__ prob, const int32_t* __restrict__ alias,
                     int n, float u1, float u2) {
  int i = (int)(u1 * (float)n);
  if (i >= n) i = n - 1;
  return (u2 < prob[i]) ? i : alias[i];
}

#endif  // __HIPCC__

}  // namespace dblink
