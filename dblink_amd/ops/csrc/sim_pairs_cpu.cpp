// CPU (OpenMP) banded Levenshtein sim-pair sweep over an attribute domain.
//
// Replaces the reference's Spark cartesian V x V sweep
// (AttributeIndex.scala:219-231) with a threshold-pruned exact pass:
// sim(a,b) > 0 requires unit(a,b) > thr/max, i.e. edit distance
// d < (|a|+|b|) * (1-u0)/(1+u0), which bounds |len(a)-len(b)| and enables a
// banded DP with early exit. Used at index-build time on the host; the GPU
// variant lives in kernels.hip (sim_pairs_gpu).

#include <torch/extension.h>

#include <algorithm>
#include <cmath>
#include <cstdint>
#include <vector>

namespace dblink {

// Edit distance with early exit once every cell of a row exceeds `band`
// (the exact distance is then irrelevant: sim truncates to 0). Attribute
// values are short (<= 64 bytes), so a plain rolling-array DP with the row-min
// cutoff is both simple and fast; the big win is the caller's length filter.
static int levenshtein_capped(const uint8_t* a, int la, const uint8_t* b, int lb,
                              int band, int* scratch) {
  if (la == 0) return lb;
  if (lb == 0) return la;
  int* prev = scratch;
  int* cur = scratch + lb + 1;
  for (int j = 0; j <= lb; ++j) prev[j] = j;
  for (int i = 1; i <= la; ++i) {
    cur[0] = i;
    int row_min = i;
    const uint8_t ca = a[i - 1];
    for (int j = 1; j <= lb; ++j) {
      const int cost = (ca == b[j - 1]) ? 0 : 1;
      int m = prev[j - 1] + cost;
      const int del = cur[j - 1] + 1;
      const int ins = prev[j] + 1;
      if (del < m) m = del;
      if (ins < m) m = ins;
      cur[j] = m;
      if (m < row_min) row_min = m;
    }
    if (row_min > band) return band + 1;
    std::swap(prev, cur);
  }
  return prev[lb];
}

std::vector<torch::Tensor> sim_pairs_cpu(torch::Tensor strs, torch::Tensor lens,
                                         double threshold, double max_sim) {
  TORCH_CHECK(strs.dim() == 2 && strs.dtype() == torch::kUInt8);
  TORCH_CHECK(lens.dtype() == torch::kInt32);
  const int64_t V = strs.size(0);
  const int64_t max_len = strs.size(1);
  const uint8_t* S = strs.data_ptr<uint8_t>();
  const int32_t* L = lens.data_ptr<int32_t>();
  const double u0 = threshold / max_sim;
  const double scale = max_sim / (max_sim - threshold);

  std::vector<std::vector<int32_t>> cols(V);
  std::vector<std::vector<float>> sims(V);

#pragma omp parallel for schedule(dynamic, 16)
  for (int64_t i = 0; i < V; ++i) {
    const uint8_t* a = S + i * max_len;
    const int la = L[i];
    std::vector<int> scratch(2 * (max_len + 1));
    for (int64_t j = 0; j < V; ++j) {
      const int lb = L[j];
      const int tot = la + lb;
      double unit;
      if (tot == 0) {
        unit = 1.0;
      } else {
        // d < dmax for sim > 0
        const double dmax_f = tot * (1.0 - u0) / (1.0 + u0);
        const int band = (int)std::ceil(dmax_f);
        if (std::abs(la - lb) >= dmax_f) continue;
        const int d = levenshtein_capped(a, la, S + j * max_len, lb, band,
                                         scratch.data());
        if (d > band) continue;
        unit = 1.0 - 2.0 * (double)d / ((double)tot + d);
      }
      const double trans = scale * (max_sim * unit - threshold);
      // inclusion criterion matches the python oracle exactly: exp(sim) > 1
      // in f64 (knife-edge sims ~1e-17 round exp() to exactly 1.0 and are
      // excluded on both sides; they would carry zero weight anyway)
      const double es = std::exp(trans);
      if (es > 1.0) {
        cols[i].push_back((int32_t)j);
        sims[i].push_back((float)es);
      }
    }
  }

  auto row_ptr = torch::zeros({V + 1}, torch::kInt64);
  int64_t* rp = row_ptr.data_ptr<int64_t>();
  for (int64_t i = 0; i < V; ++i) rp[i + 1] = rp[i] + (int64_t)cols[i].size();
  const int64_t nnz = rp[V];
  auto col = torch::empty({nnz}, torch::kInt32);
  auto expsim = torch::empty({nnz}, torch::kFloat32);
  int32_t* cp = col.data_ptr<int32_t>();
  float* sp = expsim.data_ptr<float>();
#pragma omp parallel for schedule(static)
  for (int64_t i = 0; i < V; ++i) {
    std::copy(cols[i].begin(), cols[i].end(), cp + rp[i]);
    std::copy(sims[i].begin(), sims[i].end(), sp + rp[i]);
  }
  return {row_ptr, col, expsim};
}

}  // namespace dblink
