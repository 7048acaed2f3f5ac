// Native (OpenMP) PCG-II dense link update for the CPU engine.
//
// The collapsed link update scores every record against every entity of its
// partition (GibbsUpdates.scala:363-395) — quadratic by design. The numpy
// fast path builds [Rp, Ep] weight matrices per attribute (~1 s/sweep at the
// 10k 4-partition config); this kernel computes the same per-record
// log-weights in f64 (underflow-free) with two-value const-attr logs and
// sparse sim-row adjustments, threaded over records:
//
//   lw(e) = sum_a log[(y==x)(1-th) + th*phi(x)*norm(y)*expsim(x,y)]
//
// Draws use the same per-record Philox uniforms as the numpy path
// (inverse CDF over the normalized weights).

#include <omp.h>
#include <torch/extension.h>

#include <algorithm>
#include <cmath>
#include <cstdint>
#include <vector>

namespace dblink {

torch::Tensor pcg2_link_cpu(
    torch::Tensor rec_values, torch::Tensor rec_file, torch::Tensor rec_part,
    torch::Tensor ent_values, torch::Tensor ent_ptr, torch::Tensor theta,
    torch::Tensor phi, torch::Tensor norm, torch::Tensor log_norm,
    torch::Tensor voff, torch::Tensor csr_row_ptr, torch::Tensor csr_col,
    torch::Tensor csr_expsim, torch::Tensor attr_const,
    std::vector<torch::Tensor> post_perm,   // per non-const attr: [E] i32
    std::vector<torch::Tensor> post_ptr,    // per non-const attr: [P*Va+1] i64
    torch::Tensor u_rec) {
  const int64_t R = rec_values.size(0);
  const int A = (int)rec_values.size(1);
  const int F = (int)theta.size(1);
  const int64_t P = ent_ptr.numel() - 1;
  auto out = torch::empty({R}, torch::TensorOptions().dtype(torch::kInt64));

  const int32_t* rv = rec_values.data_ptr<int32_t>();
  const int32_t* rf = rec_file.data_ptr<int32_t>();
  const int32_t* rp = rec_part.data_ptr<int32_t>();
  const int32_t* ev = ent_values.data_ptr<int32_t>();
  const int64_t* eptr = ent_ptr.data_ptr<int64_t>();
  const double* th_ = theta.data_ptr<double>();
  const double* phi_ = phi.data_ptr<double>();
  const double* nrm = norm.data_ptr<double>();
  const double* lnrm = log_norm.data_ptr<double>();
  const int64_t* vo = voff.data_ptr<int64_t>();
  const int64_t* rptr = csr_row_ptr.data_ptr<int64_t>();
  const int32_t* rcol = csr_col.data_ptr<int32_t>();
  const double* rexp = csr_expsim.data_ptr<double>();
  const uint8_t* cst = attr_const.data_ptr<uint8_t>();
  const double* u = u_rec.data_ptr<double>();
  int64_t* o = out.data_ptr<int64_t>();

  // non-const attr -> postings list index; per-attr domain size
  std::vector<int> ncidx(A, -1);
  std::vector<int64_t> Va(A);
  int nc = 0;
  for (int a = 0; a < A; ++a) {
    Va[a] = vo[a + 1] - vo[a];
    if (!cst[a]) ncidx[a] = nc++;
  }
  TORCH_CHECK((int)post_perm.size() == nc && (int)post_ptr.size() == nc,
              "postings list count mismatch");
  std::vector<const int32_t*> pp(nc);
  std::vector<const int64_t*> pq(nc);
  for (int i = 0; i < nc; ++i) {
    pp[i] = post_perm[i].data_ptr<int32_t>();
    pq[i] = post_ptr[i].data_ptr<int64_t>();
    TORCH_CHECK(post_ptr[i].numel() >= P, "postings ptr too short");
  }

  int64_t max_ep = 0;
  for (int64_t p = 0; p < P; ++p) max_ep = std::max(max_ep, eptr[p + 1] - eptr[p]);

#pragma omp parallel
  {
    std::vector<double> lw((size_t)max_ep);
#pragma omp for schedule(dynamic, 16)
    for (int64_t r = 0; r < R; ++r) {
      const int64_t p = rp[r];
      const int64_t e0 = eptr[p];
      const int64_t Ep = eptr[p + 1] - e0;
      if (Ep == 0) { o[r] = -1; continue; }
      std::fill(lw.begin(), lw.begin() + Ep, 0.0);
      const int f = rf[r];
      for (int a = 0; a < A; ++a) {
        const int32_t x = rv[r * A + a];
        if (x < 0) continue;
        const double th = th_[a * F + f];
        const double base = th * phi_[vo[a] + x];
        if (cst[a]) {
          const double la = std::log((1.0 - th) + base);
          const double ld = std::log(base);
          for (int64_t e = 0; e < Ep; ++e)
            lw[e] += (ev[(e0 + e) * A + a] == x) ? la : ld;
        } else {
          const double c1 = std::log(base);
          for (int64_t e = 0; e < Ep; ++e)
            lw[e] += c1 + lnrm[vo[a] + ev[(e0 + e) * A + a]];
          const int ai = ncidx[a];
          const int64_t* qp = pq[ai];
          const int32_t* qm = pp[ai];
          for (int64_t j = rptr[vo[a] + x]; j < rptr[vo[a] + x + 1]; ++j) {
            const int32_t c = rcol[j];
            const double fv = rexp[j];
            const double bc = base * nrm[vo[a] + c];
            const double delta =
                std::log(fv + ((c == x) ? (1.0 - th) / bc : 0.0));
            const int64_t key = p * Va[a] + c;
            for (int64_t i = qp[key]; i < qp[key + 1]; ++i)
              lw[qm[i] - e0] += delta;
          }
        }
      }
      double m = lw[0];
      for (int64_t e = 1; e < Ep; ++e) m = std::max(m, lw[e]);
      double tot = 0.0;
      for (int64_t e = 0; e < Ep; ++e) {
        const double w = std::exp(lw[e] - m);
        lw[e] = w;
        tot += w;
      }
      const double target = u[r] * tot;
      double c = 0.0;
      int64_t sel = Ep - 1;
      for (int64_t e = 0; e < Ep; ++e) {
        c += lw[e];
        if (c >= target) { sel = e; break; }
      }
      o[r] = e0 + sel;
    }
  }
  return out;
}

}  // namespace dblink

#include "common.h"

namespace dblink {

// Indexed (PCG-I / Gibbs) link update, threaded over records: candidates are
// the smallest non-distorted posting segment (GibbsUpdates.scala:398-430,
// 473-530), weights over observed-distorted non-constant attributes, draws
// by Gumbel-max with Philox keyed (seed, iteration, record gid, entity) —
// deterministic regardless of thread schedule.
std::tuple<torch::Tensor, int64_t> pcg1_link_cpu(
    torch::Tensor rec_values, torch::Tensor rec_dist, torch::Tensor rec_part,
    torch::Tensor rec_gid, torch::Tensor rec_ent_in, torch::Tensor ent_values,
    torch::Tensor ent_ptr, torch::Tensor log_norm, torch::Tensor voff,
    torch::Tensor csr_row_ptr, torch::Tensor csr_col, torch::Tensor log_expsim,
    torch::Tensor attr_const, std::vector<torch::Tensor> post_perm,
    std::vector<torch::Tensor> post_ptr, int64_t seed, int64_t iteration) {
  const int64_t R = rec_values.size(0);
  const int A = (int)rec_values.size(1);
  auto out = torch::empty({R}, torch::TensorOptions().dtype(torch::kInt64));

  const int32_t* rv = rec_values.data_ptr<int32_t>();
  const uint8_t* rd = rec_dist.data_ptr<uint8_t>();
  const int32_t* rp = rec_part.data_ptr<int32_t>();
  const int64_t* rg = rec_gid.data_ptr<int64_t>();
  const int64_t* rin = rec_ent_in.data_ptr<int64_t>();
  const int32_t* ev = ent_values.data_ptr<int32_t>();
  const int64_t* eptr = ent_ptr.data_ptr<int64_t>();
  const double* lnrm = log_norm.data_ptr<double>();
  const int64_t* vo = voff.data_ptr<int64_t>();
  const int64_t* rptr = csr_row_ptr.data_ptr<int64_t>();
  const int32_t* rcol = csr_col.data_ptr<int32_t>();
  const double* rsim = log_expsim.data_ptr<double>();
  const uint8_t* cst = attr_const.data_ptr<uint8_t>();
  int64_t* o = out.data_ptr<int64_t>();

  std::vector<const int32_t*> pp(A);
  std::vector<const int64_t*> pq(A);
  std::vector<int64_t> Va(A);
  for (int a = 0; a < A; ++a) {
    pp[a] = post_perm[a].data_ptr<int32_t>();
    pq[a] = post_ptr[a].data_ptr<int64_t>();
    Va[a] = vo[a + 1] - vo[a];
  }
  int64_t n_empty = 0;

#pragma omp parallel reduction(+ : n_empty)
  {
#pragma omp for schedule(dynamic, 32)
    for (int64_t r = 0; r < R; ++r) {
      const int64_t p = rp[r];
      uint32_t nd_mask = 0, od_mask = 0;
      for (int a = 0; a < A; ++a) {
        const int32_t x = rv[r * A + a];
        if (x < 0) continue;
        if (!rd[r * A + a]) nd_mask |= 1u << a;
        else if (!cst[a]) od_mask |= 1u << a;
      }
      int base_a = -1;
      int64_t blo = 0, bn = INT64_MAX;
      for (uint32_t m = nd_mask; m;) {
        const int a = __builtin_ctz(m);
        m &= m - 1;
        const int64_t key = p * Va[a] + rv[r * A + a];
        const int64_t lo = pq[a][key], n = pq[a][key + 1] - lo;
        if (n < bn) { bn = n; blo = lo; base_a = a; }
      }
      if (base_a < 0) {  // no observed non-distorted attr: whole partition
        blo = eptr[p];
        bn = eptr[p + 1] - blo;
      }
      const uint32_t check = base_a >= 0 ? (nd_mask & ~(1u << base_a)) : 0u;
      const uint64_t gid = (uint64_t)rg[r];
      double best = -INFINITY;
      int64_t best_e = -1;
      for (int64_t i = 0; i < bn; ++i) {
        const int64_t e = base_a >= 0 ? (int64_t)pp[base_a][blo + i] : blo + i;
        bool okc = true;
        for (uint32_t m = check; m;) {
          const int a = __builtin_ctz(m);
          m &= m - 1;
          if (ev[e * A + a] != rv[r * A + a]) { okc = false; break; }
        }
        if (!okc) continue;
        double logw = 0.0;
        for (uint32_t m = od_mask; m;) {
          const int a = __builtin_ctz(m);
          m &= m - 1;
          const int32_t y = ev[e * A + a];
          logw += lnrm[vo[a] + y];
          const int64_t row = vo[a] + rv[r * A + a];
          int64_t lo2 = rptr[row], hi2 = rptr[row + 1];
          while (lo2 < hi2) {
            const int64_t mid = (lo2 + hi2) >> 1;
            if (rcol[mid] < y) lo2 = mid + 1; else hi2 = mid;
          }
          if (lo2 < rptr[row + 1] && rcol[lo2] == y) logw += rsim[lo2];
        }
        const float u = philox_uniform((uint64_t)seed, (uint32_t)iteration, 1u,
                                       gid, (uint32_t)e);
        const double s = logw - std::log(-std::log((double)u));
        if (s > best) { best = s; best_e = e; }
      }
      if (best_e < 0) {
        ++n_empty;
        best_e = rin[r];
      }
      o[r] = best_e;
    }
  }
  return {out, n_empty};
}

}  // namespace dblink

namespace dblink {

// CPU Philox4x32-10 with the cpu_fast keying (phase folded into the key's
// high word, counter = (id_lo, id_hi, draw, iteration)) — bitwise-identical
// to cpu_fast._philox_uniform4.
static inline void philox_cpu4(uint64_t seed, uint32_t iteration, uint32_t phase,
                               uint64_t id, uint32_t draw, int rank, double u[4]) {
  uint32_t c0 = (uint32_t)id, c1 = (uint32_t)(id >> 32), c2 = draw, c3 = iteration;
  uint32_t k0 = (uint32_t)seed;
  uint32_t k1 = (uint32_t)(seed >> 32) ^ (uint32_t)(phase * 0x9E3779B1u) ^
                (uint32_t)((uint32_t)rank * 0x85EBCA6Bu);
  for (int r = 0; r < 10; ++r) {
    const uint64_t p0 = (uint64_t)c0 * 0xD2511F53ull;
    const uint64_t p1 = (uint64_t)c2 * 0xCD9E8D57ull;
    const uint32_t n0 = (uint32_t)(p1 >> 32) ^ c1 ^ k0;
    const uint32_t n1 = (uint32_t)p1;
    const uint32_t n2 = (uint32_t)(p0 >> 32) ^ c3 ^ k1;
    const uint32_t n3 = (uint32_t)p0;
    c0 = n0; c1 = n1; c2 = n2; c3 = n3;
    k0 += 0x9E3779B9u; k1 += 0xBB67AE85u;
  }
  const double s = ldexp(1.0, -32);
  u[0] = ((double)c0 + 0.5) * s;
  u[1] = ((double)c1 + 0.5) * s;
  u[2] = ((double)c2 + 0.5) * s;
  u[3] = ((double)c3 + 0.5) * s;
}

static inline int alias_draw_cpu(double u1, double u2, const double* prob,
                                 const int64_t* alias, int64_t n) {
  int64_t idx = (int64_t)(u1 * (double)n);
  if (idx > n - 1) idx = n - 1;
  return (int)(u2 < prob[idx] ? idx : alias[idx]);
}

// The collapsed / non-collapsed entity-value update (K6,
// GibbsUpdates.scala:576-727), threaded over entities. Bitwise-identical to
// the numpy fast path (same Philox streams, same f64 op order); pairs with
// k_obs beyond the cached power tables are returned for the python oracle
// fallback.
std::tuple<torch::Tensor, torch::Tensor> value_update_cpu(
    torch::Tensor rec_values, torch::Tensor rec_file, torch::Tensor rec_dist,
    torch::Tensor rec_ent, torch::Tensor rorder, torch::Tensor starts,
    torch::Tensor ent_values, torch::Tensor theta, torch::Tensor phi,
    torch::Tensor norm, torch::Tensor voff, torch::Tensor csr_row_ptr,
    torch::Tensor csr_col, torch::Tensor csr_expsim, torch::Tensor attr_const,
    std::vector<torch::Tensor> phi_prob, std::vector<torch::Tensor> phi_alias,
    std::vector<torch::Tensor> pow_prob, std::vector<torch::Tensor> pow_alias,
    torch::Tensor pow_totals, int64_t kmax, int64_t collapsed, int64_t seed,
    int64_t iteration, int64_t rank) {
  const int64_t E = ent_values.size(0);
  const int A = (int)ent_values.size(1);
  const int F = (int)theta.size(1);
  auto new_ev = ent_values.clone();
  const int32_t* rv = rec_values.data_ptr<int32_t>();
  const int32_t* rf = rec_file.data_ptr<int32_t>();
  const uint8_t* rd = rec_dist.data_ptr<uint8_t>();
  const int64_t* ro = rorder.data_ptr<int64_t>();
  const int64_t* st = starts.data_ptr<int64_t>();
  int32_t* ev = new_ev.data_ptr<int32_t>();
  const double* th_ = theta.data_ptr<double>();
  const double* phi_ = phi.data_ptr<double>();
  const double* nrm = norm.data_ptr<double>();
  const int64_t* vo = voff.data_ptr<int64_t>();
  const int64_t* rptr = csr_row_ptr.data_ptr<int64_t>();
  const int32_t* rcol = csr_col.data_ptr<int32_t>();
  const double* rexp = csr_expsim.data_ptr<double>();
  const uint8_t* cst = attr_const.data_ptr<uint8_t>();
  const double* ptot = pow_totals.data_ptr<double>();  // [A, kmax+1]
  std::vector<const double*> php(A), pwp(A);
  std::vector<const int64_t*> pha(A), pwa(A);
  for (int a = 0; a < A; ++a) {
    php[a] = phi_prob[a].data_ptr<double>();
    pha[a] = phi_alias[a].data_ptr<int64_t>();
    pwp[a] = pow_prob[a].numel() ? pow_prob[a].data_ptr<double>() : nullptr;
    pwa[a] = pow_alias[a].numel() ? pow_alias[a].data_ptr<int64_t>() : nullptr;
  }
  std::vector<std::vector<int64_t>> fb_per_thread(64);

#pragma omp parallel
  {
    const int tid = omp_get_thread_num() & 63;
    std::vector<std::pair<int32_t, double>> buf;   // (col, factor) entries
    std::vector<double> cum;
#pragma omp for schedule(dynamic, 64)
    for (int64_t e = 0; e < E; ++e) {
      for (int a = 0; a < A; ++a) {
        const int64_t V = vo[a + 1] - vo[a];
        double u[4];
        philox_cpu4((uint64_t)seed, (uint32_t)iteration, 3u,
                    (uint64_t)(e * A + a), 0u, (int)rank, u);
        const double u_mix = u[0], u_a1 = u[1], u_a2 = u[2], u_sel = u[3];
        // gather observed linked records for this attribute
        int k = 0;
        int32_t pinned = -1;
        buf.clear();
        for (int64_t j = st[e]; j < st[e + 1]; ++j) {
          const int64_t r = ro[j];
          const int32_t x = rv[r * A + a];
          if (x < 0) continue;
          ++k;
          if (!collapsed && !rd[r * A + a]) pinned = x;  // last write wins
          buf.emplace_back(x, (double)rf[r]);  // (value, file) of obs records
        }
        if (!collapsed && pinned >= 0) { ev[e * A + a] = pinned; continue; }
        int keff = k;
        if (!collapsed && cst[a] && keff >= 1) keff = 0;  // plain phi draw
        if (keff == 0) {
          ev[e * A + a] = alias_draw_cpu(u_a1, u_a2, php[a], pha[a], V);
          continue;
        }
        if (keff == 1) {
          const int32_t x = buf[0].first;
          const double th = th_[a * F + (int)buf[0].second];
          if (cst[a]) {  // collapsed closed form: P(base) = theta
            ev[e * A + a] = u_mix < th
                ? alias_draw_cpu(u_a1, u_a2, php[a], pha[a], V)
                : x;
            continue;
          }
          const double Z1 = ptot[a * (kmax + 1) + 1];
          const int64_t lo = rptr[vo[a] + x], hi = rptr[vo[a] + x + 1];
          cum.clear();
          double c = 0.0;
          for (int64_t j = lo; j < hi; ++j) {
            double w = rexp[j];
            if (collapsed && rcol[j] == x)
              w += (1.0 / th - 1.0) / (phi_[vo[a] + x] * nrm[vo[a] + x]);
            const double wgt =
                (phi_[vo[a] + rcol[j]] * nrm[vo[a] + rcol[j]] / Z1) * (w - 1.0);
            c += wgt;
            cum.push_back(c);
          }
          const double tot = c;
          if (u_mix < 1.0 / (1.0 + tot)) {
            ev[e * A + a] = alias_draw_cpu(u_a1, u_a2, pwp[a], pwa[a], V);
          } else {
            const double target = u_sel * tot;
            int64_t j = std::upper_bound(cum.begin(), cum.end(), target) -
                        cum.begin();
            if (j > (int64_t)cum.size() - 1) j = (int64_t)cum.size() - 1;
            ev[e * A + a] = rcol[lo + j];
          }
          continue;
        }
        if (keff > kmax) {  // beyond cached powers: python oracle fallback
          fb_per_thread[tid].push_back(e * A + a);
          continue;
        }
        // ---- k >= 2: union-combine the records' factor rows --------------
        // entries in record order; stable sort by col keeps the product
        // order identical to numpy's lexsort + multiply.reduceat
        const int64_t n_obs = (int64_t)buf.size();
        std::vector<std::pair<int32_t, double>> ents;
        for (int64_t m = 0; m < n_obs; ++m) {
          const int32_t x = buf[m].first;
          const double th = th_[a * F + (int)buf[m].second];
          if (cst[a]) {
            ents.emplace_back(x, 1.0 + (1.0 / th - 1.0) / phi_[vo[a] + x]);
          } else {
            const int64_t lo = rptr[vo[a] + x], hi = rptr[vo[a] + x + 1];
            for (int64_t j = lo; j < hi; ++j) {
              double w = rexp[j];
              if (collapsed && rcol[j] == x)
                w += (1.0 / th - 1.0) / (phi_[vo[a] + x] * nrm[vo[a] + x]);
              ents.emplace_back(rcol[j], w);
            }
          }
        }
        std::stable_sort(ents.begin(), ents.end(),
                         [](const std::pair<int32_t, double>& p,
                            const std::pair<int32_t, double>& q) {
                           return p.first < q.first;
                         });
        // run-reduce products per distinct col, then weights + CDF
        cum.clear();
        std::vector<int32_t> ucol;
        const double Zk = cst[a] ? 1.0 : ptot[a * (kmax + 1) + keff];
        double c = 0.0;
        for (size_t i = 0; i < ents.size();) {
          const int32_t v = ents[i].first;
          double vw = ents[i].second;
          for (++i; i < ents.size() && ents[i].first == v; ++i) vw *= ents[i].second;
          double basep;
          if (cst[a]) basep = phi_[vo[a] + v];
          else
            basep = phi_[vo[a] + v] * std::pow(nrm[vo[a] + v], (double)keff) / Zk;
          c += basep * (vw - 1.0);
          cum.push_back(c);
          ucol.push_back(v);
        }
        const double tot = c;
        if (u_mix < 1.0 / (1.0 + tot)) {
          if (cst[a])
            ev[e * A + a] = alias_draw_cpu(u_a1, u_a2, php[a], pha[a], V);
          else
            ev[e * A + a] = alias_draw_cpu(
                u_a1, u_a2, pwp[a] + (int64_t)(keff - 1) * V,
                pwa[a] + (int64_t)(keff - 1) * V, V);
        } else {
          const double target = u_sel * tot;
          int64_t j = std::upper_bound(cum.begin(), cum.end(), target) -
                      cum.begin();
          if (j > (int64_t)cum.size() - 1) j = (int64_t)cum.size() - 1;
          ev[e * A + a] = ucol[j];
        }
      }
    }
  }
  std::vector<int64_t> fb;
  for (auto& v : fb_per_thread) fb.insert(fb.end(), v.begin(), v.end());
  std::sort(fb.begin(), fb.end());
  auto fbt = torch::empty({(int64_t)fb.size()},
                          torch::TensorOptions().dtype(torch::kInt64));
  std::copy(fb.begin(), fb.end(), fbt.data_ptr<int64_t>());
  return {new_ev, fbt};
}

}  // namespace dblink

namespace dblink {

// Stable counting argsort over small-range non-negative int keys: the CPU
// fast path re-sorts entities/records by (partition, value)-style keys every
// sweep, and numpy's stable mergesort was ~25% of the real-RLdata10000
// stationary sweep. O(n + k), identical permutation to
// np.argsort(kind="stable").
torch::Tensor counting_argsort_cpu(torch::Tensor keys, int64_t k) {
  TORCH_CHECK(keys.dtype() == torch::kInt64, "counting_argsort: int64 keys");
  TORCH_CHECK(keys.is_contiguous());
  const int64_t n = keys.numel();
  auto out = torch::empty({n}, torch::kInt64);
  const int64_t* kp = keys.data_ptr<int64_t>();
  int64_t* op = out.data_ptr<int64_t>();
  std::vector<int64_t> ptr((size_t)k + 1, 0);
  for (int64_t i = 0; i < n; ++i) {
    TORCH_CHECK(kp[i] >= 0 && kp[i] < k, "counting_argsort: key out of range");
    ++ptr[(size_t)kp[i] + 1];
  }
  for (int64_t v = 0; v < k; ++v) ptr[(size_t)v + 1] += ptr[(size_t)v];
  for (int64_t i = 0; i < n; ++i) op[ptr[(size_t)kp[i]]++] = i;
  return out;
}

}  // namespace dblink

namespace dblink {

// K7 distortion resample (GibbsUpdates.scala:324-359) for the CPU fast
// path: bitwise-identical to the numpy phase (same Philox stream — packed
// four positions per counter like cpu_fast._philox_dense — and the same
// f64 expressions).
torch::Tensor distortion_update_cpu(
    torch::Tensor rec_values, torch::Tensor rec_file, torch::Tensor rec_ent,
    torch::Tensor ent_values, torch::Tensor theta, torch::Tensor self_mass,
    torch::Tensor voff, int64_t seed, int64_t iteration, int64_t rank) {
  const int64_t R = rec_values.size(0);
  const int A = (int)rec_values.size(1);
  const int64_t F = theta.size(1);
  auto out = torch::empty({R, (int64_t)A}, torch::kUInt8);
  const int32_t* rv = rec_values.data_ptr<int32_t>();
  const int32_t* rf = rec_file.data_ptr<int32_t>();
  const int64_t* re = rec_ent.data_ptr<int64_t>();
  const int32_t* ev = ent_values.data_ptr<int32_t>();
  const double* th = theta.data_ptr<double>();
  const double* sm = self_mass.data_ptr<double>();
  const int64_t* vo = voff.data_ptr<int64_t>();
  uint8_t* z = out.data_ptr<uint8_t>();
  const int64_t n = R * (int64_t)A;
  const int64_t blocks = (n + 3) / 4;
#pragma omp parallel for schedule(static)
  for (int64_t blk = 0; blk < blocks; ++blk) {
    double u[4];
    philox_cpu4((uint64_t)seed, (uint32_t)iteration, 2u, (uint64_t)blk, 0u,
                (int)rank, u);
    const int64_t lim = std::min<int64_t>(4, n - blk * 4);
    for (int64_t w = 0; w < lim; ++w) {
      const int64_t p = blk * 4 + w;
      const int64_t r = p / A;
      const int a = (int)(p % A);
      const int32_t x = rv[p];
      const double ta = th[(int64_t)a * F + rf[r]];
      bool zd;
      if (x < 0) {
        zd = u[w] < ta;
      } else if (x == ev[re[r] * A + a]) {
        const double pr1 = ta * sm[vo[a] + x];
        zd = u[w] < pr1 / (pr1 + (1.0 - ta));
      } else {
        zd = true;
      }
      z[p] = zd ? 1 : 0;
    }
  }
  return out;
}

}  // namespace dblink

namespace dblink {

// K8 summary reduction (GibbsUpdates.scala:219-301) for the CPU engine:
// log-likelihood + isolates + per-(attr, file) distortion counts + the
// per-record distortion histogram in one OpenMP pass. Matches the numpy
// compute_summary term-for-term; float summation ORDER differs (OMP
// reduction vs numpy pairwise), so the log-likelihood may differ in ulps —
// every consumer treats it as a diagnostic series.
std::tuple<double, int64_t, torch::Tensor, torch::Tensor> summary_cpu(
    torch::Tensor rec_values, torch::Tensor rec_dist, torch::Tensor rec_file,
    torch::Tensor rec_ent, torch::Tensor ent_values, torch::Tensor probs,
    torch::Tensor log_probs, torch::Tensor sim_norms, torch::Tensor voff,
    torch::Tensor csr_row_ptr, torch::Tensor csr_col, torch::Tensor csr_expsim,
    torch::Tensor attr_const, int64_t num_files) {
  const int64_t R = rec_values.size(0);
  const int A = (int)rec_values.size(1);
  const int64_t E = ent_values.size(0);
  const int64_t F = num_files;
  const int32_t* rv = rec_values.data_ptr<int32_t>();
  const uint8_t* rd = rec_dist.data_ptr<uint8_t>();
  const int32_t* rf = rec_file.data_ptr<int32_t>();
  const int64_t* re = rec_ent.data_ptr<int64_t>();
  const int32_t* ev = ent_values.data_ptr<int32_t>();
  const double* ph = probs.data_ptr<double>();
  const double* lp = log_probs.data_ptr<double>();
  const double* sn = sim_norms.data_ptr<double>();
  const int64_t* vo = voff.data_ptr<int64_t>();
  const int64_t* rp = csr_row_ptr.data_ptr<int64_t>();
  const int32_t* cc = csr_col.data_ptr<int32_t>();
  const double* ce = csr_expsim.data_ptr<double>();
  const uint8_t* cst = attr_const.data_ptr<uint8_t>();

  auto agg = torch::zeros({(int64_t)A, F}, torch::kInt64);
  auto hist = torch::zeros({(int64_t)A + 1}, torch::kInt64);
  int64_t* aggp = agg.data_ptr<int64_t>();
  int64_t* hp = hist.data_ptr<int64_t>();

  // per-thread partials summed in THREAD ORDER afterwards: the reduction
  // must be deterministic run-to-run (diagnostics.csv reproducibility) —
  // an omp critical / reduction clause would sum in arrival order
  const int nthr = omp_get_max_threads();
  std::vector<double> ll_ent(nthr, 0.0), ll_rec(nthr, 0.0);
#pragma omp parallel
  {
    const int tid = omp_get_thread_num();
    double ll = 0.0;
#pragma omp for schedule(static)
    for (int64_t e = 0; e < E; ++e)
      for (int a = 0; a < A; ++a) ll += lp[vo[a] + ev[e * A + a]];
    ll_ent[tid] = ll;
  }

  std::vector<uint8_t> linked(E, 0);
  for (int64_t r = 0; r < R; ++r) linked[re[r]] = 1;
  int64_t iso = 0;
  for (int64_t e = 0; e < E; ++e) iso += linked[e] == 0;

#pragma omp parallel
  {
    const int tid = omp_get_thread_num();
    std::vector<int64_t> agg_l((size_t)A * F, 0);
    std::vector<int64_t> hist_l((size_t)A + 1, 0);
    double ll = 0.0;
#pragma omp for schedule(static) nowait
    for (int64_t r = 0; r < R; ++r) {
      int nd = 0;
      for (int a = 0; a < A; ++a) {
        if (!rd[r * A + a]) continue;
        ++nd;
        ++agg_l[(size_t)a * F + rf[r]];
        const int32_t x = rv[r * A + a];
        if (x < 0) continue;
        const int64_t row = vo[a] + x;
        if (cst[a]) {  // constant attr: term = log(phi(x))
          ll += lp[row];
          continue;
        }
        const int32_t y = ev[re[r] * A + a];
        // expsim(x, y): binary search x's row; absent pairs have expsim 1
        double es = 1.0;
        int64_t lo = rp[row], hi = rp[row + 1];
        while (lo < hi) {
          const int64_t mid = (lo + hi) >> 1;
          if (cc[mid] < y)
            lo = mid + 1;
          else
            hi = mid;
        }
        if (lo < rp[row + 1] && cc[lo] == y) es = ce[lo];
        ll += std::log(ph[row] * sn[vo[a] + y] * es);
      }
      ++hist_l[nd];
    }
    ll_rec[tid] = ll;
#pragma omp critical
    {
      for (size_t i = 0; i < agg_l.size(); ++i) aggp[i] += agg_l[i];
      for (size_t i = 0; i < hist_l.size(); ++i) hp[i] += hist_l[i];
    }
  }
  double loglik = 0.0;
  for (int t = 0; t < nthr; ++t) loglik += ll_ent[t];
  for (int t = 0; t < nthr; ++t) loglik += ll_rec[t];
  return {loglik, iso, agg, hist};
}

}  // namespace dblink

namespace dblink {

// MPC/sMPC integer core helpers (analysis/chain.py): order-independent
// cluster content keys, composite sort keys and first-occurrence — all
// integer arithmetic, bitwise-identical to the numpy expressions they
// replace (mod-2^64 sums commute).

static inline uint64_t splitmix64(uint64_t x) {
  x += 0x9E3779B97F4A7C15ull;
  x ^= x >> 30;
  x *= 0xBF58476D1CE4E5B9ull;
  x ^= x >> 27;
  x *= 0x94D049BB133111EBull;
  x ^= x >> 31;
  return x;
}

torch::Tensor mpc_cluster_keys(torch::Tensor codes, torch::Tensor offsets) {
  const int64_t C = offsets.numel() - 1;
  auto out = torch::empty({C}, torch::kInt64);
  const int32_t* cp = codes.data_ptr<int32_t>();
  const int64_t* op = offsets.data_ptr<int64_t>();
  int64_t* kp = out.data_ptr<int64_t>();
#pragma omp parallel for schedule(static)
  for (int64_t c = 0; c < C; ++c) {
    uint64_t s1 = 0, s2 = 0;
    for (int64_t j = op[c]; j < op[c + 1]; ++j) {
      const uint64_t h1 = splitmix64((uint64_t)(uint32_t)cp[j]);
      uint64_t h2 = (h1 ^ (h1 >> 29)) * 0xD6E8FEB86659FD93ull;
      h2 ^= h2 >> 32;
      s1 += h1;
      s2 += h2;
    }
    const uint64_t key =
        (s1 ^ (s2 * 0x9E3779B97F4A7C15ull)) + (uint64_t)(op[c + 1] - op[c]);
    kp[c] = (int64_t)key;
  }
  return out;
}

torch::Tensor mpc_combo(torch::Tensor codes, torch::Tensor offsets,
                        torch::Tensor cluster_counts, int64_t cnt_bits,
                        int64_t idx_bits) {
  const int64_t C = offsets.numel() - 1;
  const int64_t n = codes.numel();
  auto out = torch::empty({n}, torch::kInt64);
  const int32_t* cp = codes.data_ptr<int32_t>();
  const int64_t* op = offsets.data_ptr<int64_t>();
  const int64_t* cc = cluster_counts.data_ptr<int64_t>();
  int64_t* ob = out.data_ptr<int64_t>();
#pragma omp parallel for schedule(static)
  for (int64_t c = 0; c < C; ++c) {
    const int64_t cnt_sh = cc[c] << idx_bits;
    for (int64_t j = op[c]; j < op[c + 1]; ++j)
      ob[j] = ((int64_t)cp[j] << (cnt_bits + idx_bits)) | cnt_sh | (n - 1 - j);
  }
  return out;
}

torch::Tensor first_occurrence(torch::Tensor kcode, int64_t k) {
  const int64_t C = kcode.numel();
  auto out = torch::full({k}, -1, torch::kInt64);
  const int64_t* kp = kcode.data_ptr<int64_t>();
  int64_t* op = out.data_ptr<int64_t>();
  for (int64_t i = 0; i < C; ++i)
    if (op[kp[i]] < 0) op[kp[i]] = i;
  return out;
}

}  // namespace dblink
