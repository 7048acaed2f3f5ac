// Python bindings for the dblink_amd native extension.

#include <torch/extension.h>

#include <vector>

namespace dblink {

// link_dense_cpu.cpp
torch::Tensor pcg2_link_cpu(torch::Tensor rec_values, torch::Tensor rec_file,
                            torch::Tensor rec_part, torch::Tensor ent_values,
                            torch::Tensor ent_ptr, torch::Tensor theta,
                            torch::Tensor phi, torch::Tensor norm,
                            torch::Tensor log_norm, torch::Tensor voff,
                            torch::Tensor csr_row_ptr, torch::Tensor csr_col,
                            torch::Tensor csr_expsim, torch::Tensor attr_const,
                            std::vector<torch::Tensor> post_perm,
                            std::vector<torch::Tensor> post_ptr,
                            torch::Tensor u_rec);

std::tuple<torch::Tensor, int64_t> pcg1_link_cpu(
    torch::Tensor rec_values, torch::Tensor rec_dist, torch::Tensor rec_part,
    torch::Tensor rec_gid, torch::Tensor rec_ent_in, torch::Tensor ent_values,
    torch::Tensor ent_ptr, torch::Tensor log_norm, torch::Tensor voff,
    torch::Tensor csr_row_ptr, torch::Tensor csr_col, torch::Tensor log_expsim,
    torch::Tensor attr_const, std::vector<torch::Tensor> post_perm,
    std::vector<torch::Tensor> post_ptr, int64_t seed, int64_t iteration);

torch::Tensor counting_argsort_cpu(torch::Tensor keys, int64_t k);

torch::Tensor mpc_cluster_keys(torch::Tensor codes, torch::Tensor offsets);

torch::Tensor mpc_combo(torch::Tensor codes, torch::Tensor offsets,
                        torch::Tensor cluster_counts, int64_t cnt_bits,
                        int64_t idx_bits);

torch::Tensor first_occurrence(torch::Tensor kcode, int64_t k);

std::tuple<double, int64_t, torch::Tensor, torch::Tensor> summary_cpu(
    torch::Tensor rec_values, torch::Tensor rec_dist, torch::Tensor rec_file,
    torch::Tensor rec_ent, torch::Tensor ent_values, torch::Tensor probs,
    torch::Tensor log_probs, torch::Tensor sim_norms, torch::Tensor voff,
    torch::Tensor csr_row_ptr, torch::Tensor csr_col, torch::Tensor csr_expsim,
    torch::Tensor attr_const, int64_t num_files);

torch::Tensor distortion_update_cpu(
    torch::Tensor rec_values, torch::Tensor rec_file, torch::Tensor rec_ent,
    torch::Tensor ent_values, torch::Tensor theta, torch::Tensor self_mass,
    torch::Tensor voff, int64_t seed, int64_t iteration, int64_t rank);

std::tuple<torch::Tensor, torch::Tensor> value_update_cpu(
    torch::Tensor rec_values, torch::Tensor rec_file, torch::Tensor rec_dist,
    torch::Tensor rec_ent, torch::Tensor rorder, torch::Tensor starts,
    torch::Tensor ent_values, torch::Tensor theta, torch::Tensor phi,
    torch::Tensor norm, torch::Tensor voff, torch::Tensor csr_row_ptr,
    torch::Tensor csr_col, torch::Tensor csr_expsim, torch::Tensor attr_const,
    std::vector<torch::Tensor> phi_prob, std::vector<torch::Tensor> phi_alias,
    std::vector<torch::Tensor> pow_prob, std::vector<torch::Tensor> pow_alias,
    torch::Tensor pow_totals, int64_t kmax, int64_t collapsed, int64_t seed,
    int64_t iteration, int64_t rank);

// sim_pairs_cpu.cpp
std::vector<torch::Tensor> sim_pairs_cpu(torch::Tensor strs, torch::Tensor lens,
                                         double threshold, double max_sim);

// kernels.hip
void link_update(torch::Tensor rec_values, torch::Tensor rec_dist,
                 torch::Tensor rec_gid, torch::Tensor rec_part,
                 torch::Tensor cand_lo, torch::Tensor cand_hi,
                 torch::Tensor postings, torch::Tensor ent_values,
                 torch::Tensor ent_ptr, torch::Tensor log_norm, torch::Tensor voff,
                 torch::Tensor csr_row_ptr, torch::Tensor csr_col,
                 torch::Tensor csr_sim, torch::Tensor attr_const, int64_t seed,
                 int64_t iteration, torch::Tensor rec_ent_out,
                 torch::Tensor rec_ent_in, torch::Tensor error_count,
                 torch::Tensor small_mask, torch::Tensor ctrl,
                 torch::Tensor pair_a1, torch::Tensor pair_a2);
void postings_hist(torch::Tensor ent_part, torch::Tensor ent_values,
                   torch::Tensor pair_a1, torch::Tensor pair_a2,
                   torch::Tensor pair_v2, int64_t Vmax, torch::Tensor counts);
void postings_scatter(torch::Tensor ent_part, torch::Tensor ent_values,
                      torch::Tensor pair_a1, torch::Tensor pair_a2,
                      torch::Tensor pair_v2, int64_t Vmax, torch::Tensor cursor,
                      torch::Tensor postings);
void cand_ranges(torch::Tensor rec_part, torch::Tensor rec_values,
                 torch::Tensor pair_a1, torch::Tensor pair_a2,
                 torch::Tensor pair_v2, torch::Tensor ptr, int64_t Vmax,
                 torch::Tensor cand_lo, torch::Tensor cand_hi);
void classify_modes(torch::Tensor rec_values, torch::Tensor rec_dist,
                    torch::Tensor rec_part, torch::Tensor ent_ptr,
                    torch::Tensor cand_lo, torch::Tensor cand_hi, int64_t NP,
                    int64_t small_threshold, int64_t heavy_threshold,
                    torch::Tensor mode);
void build_ekeys_stable(torch::Tensor ent_part, torch::Tensor ent_values,
                        torch::Tensor pair_a1, torch::Tensor pair_a2,
                        torch::Tensor pair_v2, int64_t Vmax, torch::Tensor ekeys);
int64_t radix_sort_pairs_temp_bytes(int64_t n);
void radix_sort_pairs_i64_i32(torch::Tensor keys_in, torch::Tensor keys_out,
                              torch::Tensor vals_in, torch::Tensor vals_out,
                              int64_t end_bit, torch::Tensor temp);
void link_update_heavy(torch::Tensor mode, torch::Tensor rec_values,
                       torch::Tensor rec_dist, torch::Tensor rec_gid,
                       torch::Tensor rec_part, torch::Tensor ent_values,
                       torch::Tensor ent_ptr, torch::Tensor log_norm,
                       torch::Tensor voff, torch::Tensor csr_row_ptr,
                       torch::Tensor csr_col, torch::Tensor csr_sim,
                       torch::Tensor attr_const, torch::Tensor csr_row_ptr_big,
                       torch::Tensor csr_col_big, torch::Tensor csr_sim_big,
                       double tau, torch::Tensor postings,
                       torch::Tensor idx_ptr, int64_t Vmax, int64_t NP,
                       int64_t seed, int64_t iteration, torch::Tensor ctrl,
                       torch::Tensor rec_ent_out, torch::Tensor rec_ent_in,
                       torch::Tensor error_count, torch::Tensor stats);
void link_update_dense(torch::Tensor rec_values, torch::Tensor rec_dist,
                       torch::Tensor rec_gid, torch::Tensor rec_part,
                       torch::Tensor rec_file, torch::Tensor ent_values,
                       torch::Tensor ent_ptr, torch::Tensor theta, torch::Tensor phi,
                       torch::Tensor norm_lin, torch::Tensor voff,
                       torch::Tensor csr_row_ptr, torch::Tensor csr_col,
                       torch::Tensor csr_sim, torch::Tensor attr_const,
                       int64_t collapsed, int64_t seed, int64_t iteration,
                       torch::Tensor rec_ent_out, torch::Tensor ctrl);
void value_update(torch::Tensor rec_values, torch::Tensor rec_dist,
                  torch::Tensor rec_file, torch::Tensor ent_rec_ptr,
                  torch::Tensor ent_rec_idx, torch::Tensor ent_values,
                  torch::Tensor theta, torch::Tensor phi, torch::Tensor log_phi,
                  torch::Tensor norm_lin, torch::Tensor log_norm, torch::Tensor voff,
                  torch::Tensor csr_row_ptr, torch::Tensor csr_col,
                  torch::Tensor csr_sim, torch::Tensor phi_prob,
                  torch::Tensor phi_alias, torch::Tensor pow_prob,
                  torch::Tensor pow_alias, torch::Tensor pow_off,
                  torch::Tensor log_pow_total, torch::Tensor attr_const, int64_t Kc,
                  int64_t collapsed, int64_t sequential, int64_t seed,
                  int64_t iteration, int64_t ent_id_base, torch::Tensor error_count,
                  torch::Tensor wave_pairs, torch::Tensor base_pairs,
                  torch::Tensor k1_pairs, torch::Tensor csr_excl,
                  torch::Tensor csr_rawsum, torch::Tensor z1, torch::Tensor ctrl,
                  torch::Tensor kobs);
void distortion_update(torch::Tensor rec_values, torch::Tensor rec_dist,
                       torch::Tensor rec_file, torch::Tensor rec_gid,
                       torch::Tensor rec_ent, torch::Tensor ent_values,
                       torch::Tensor theta, torch::Tensor phi,
                       torch::Tensor norm_lin, torch::Tensor self_expsim,
                       torch::Tensor voff, torch::Tensor attr_const, int64_t seed,
                       int64_t iteration, torch::Tensor ctrl, torch::Tensor log_phi,
                       torch::Tensor log_norm, torch::Tensor csr_row_ptr,
                       torch::Tensor csr_col, torch::Tensor csr_sim,
                       torch::Tensor loglik);
void summary_loglik(torch::Tensor ent_values, torch::Tensor rec_values,
                    torch::Tensor rec_dist, torch::Tensor rec_ent,
                    torch::Tensor log_phi, torch::Tensor log_norm, torch::Tensor voff,
                    torch::Tensor csr_row_ptr, torch::Tensor csr_col,
                    torch::Tensor csr_sim, torch::Tensor attr_const,
                    torch::Tensor out);
void build_keys(torch::Tensor ent_part, torch::Tensor ent_values,
                torch::Tensor rec_part, torch::Tensor rec_values,
                torch::Tensor pair_a1, torch::Tensor pair_a2,
                torch::Tensor pair_v2, int64_t Vmax,
                torch::Tensor ekeys, torch::Tensor qkeys);
void set_value_stats(torch::Tensor t);
void set_value_ktables(torch::Tensor excl, torch::Tensor rawsum,
                       torch::Tensor self_expsim, int64_t kmax, int64_t nnz);
void set_value_k2tables(torch::Tensor excl, torch::Tensor rawsum, int64_t k2max);
void mfma_score_bench(torch::Tensor rcode, torch::Tensor rbonus,
                      torch::Tensor ecode, int64_t K, torch::Tensor score);
void scalar_score_bench(torch::Tensor rcode, torch::Tensor rbonus,
                        torch::Tensor ecode, torch::Tensor score);
void summary_counts(torch::Tensor rec_dist, torch::Tensor rec_file,
                    torch::Tensor ent_rec_ptr, int64_t E, torch::Tensor counts,
                    torch::Tensor loglik, torch::Tensor packed);
void kd_descent(torch::Tensor ent_values, torch::Tensor node_kind,
                torch::Tensor node_attr, torch::Tensor node_a, torch::Tensor node_b,
                torch::Tensor rset, torch::Tensor ent_part_out);
std::vector<torch::Tensor> sim_pairs_gpu(torch::Tensor strs, torch::Tensor lens,
                                         double threshold, double max_sim);

}  // namespace dblink

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "dblink_amd native ops (CDNA4 HIP kernels + host helpers)";
  m.def("value_update_cpu", &dblink::value_update_cpu,
        "entity-value update (OpenMP, bitwise-matches the numpy fast path)");
  m.def("pcg1_link_cpu", &dblink::pcg1_link_cpu,
        "indexed PCG-I/Gibbs link update (OpenMP, keyed Philox gumbels)");
  m.def("pcg2_link_cpu", &dblink::pcg2_link_cpu,
        "PCG-II dense link update (OpenMP, f64 log-space)");
  m.def("counting_argsort_cpu", &dblink::counting_argsort_cpu,
        "stable counting argsort for small-range int64 keys");
  m.def("mpc_cluster_keys", &dblink::mpc_cluster_keys,
        "order-independent cluster content hashes (MPC/sMPC core)");
  m.def("mpc_combo", &dblink::mpc_combo,
        "composite (code | count | inv index) sort keys (MPC core)");
  m.def("first_occurrence", &dblink::first_occurrence,
        "first index of each key value");
  m.def("distortion_update_cpu", &dblink::distortion_update_cpu,
        "distortion resample (OpenMP, bitwise-matches the numpy fast path)");
  m.def("summary_cpu", &dblink::summary_cpu,
        "summary reduction: loglik + isolates + distortion counts (OpenMP)");
  m.def("sim_pairs_cpu", &dblink::sim_pairs_cpu,
        "banded Levenshtein sim-pair sweep (CPU/OpenMP)");
  m.def("sim_pairs_gpu", &dblink::sim_pairs_gpu,
        "banded Levenshtein sim-pair sweep (gfx950)");
  m.def("link_update", &dblink::link_update, "K3/K4/K5 fused link update");
  m.def("link_update_dense", &dblink::link_update_dense,
        "dense link update (PCG-II / Gibbs-Sequential)");
  m.def("value_update", &dblink::value_update, "K6 entity-value update");
  m.def("distortion_update", &dblink::distortion_update, "K7 distortion resample");
  m.def("summary_loglik", &dblink::summary_loglik, "K8 log-likelihood reduction");
  m.def("kd_descent", &dblink::kd_descent, "K9a KD-tree partition reassignment");
  m.def("build_keys", &dblink::build_keys, "fused inverted-index key build");
  m.def("postings_hist", &dblink::postings_hist,
        "counting-sort index: per-key histogram");
  m.def("postings_scatter", &dblink::postings_scatter,
        "counting-sort index: posting scatter");
  m.def("cand_ranges", &dblink::cand_ranges,
        "per-record candidate ranges from the dense key prefix");
  m.def("classify_modes", &dblink::classify_modes,
        "route records to the wave / thread / hierarchical link paths");
  m.def("build_ekeys_stable", &dblink::build_ekeys_stable,
        "stable inverted-index sort keys (key * E + entity)");
  m.def("radix_sort_pairs_temp_bytes", &dblink::radix_sort_pairs_temp_bytes,
        "rocprim radix-sort workspace size");
  m.def("radix_sort_pairs_i64_i32", &dblink::radix_sort_pairs_i64_i32,
        "hipGraph-safe rocprim radix sort (persistent workspace)");
  m.def("link_update_heavy", &dblink::link_update_heavy,
        "hierarchical (A*) Gumbel-max link update for huge candidate sets");
  m.def("summary_counts", &dblink::summary_counts, "fused summary counts + pack");
  m.def("set_value_stats", &dblink::set_value_stats,
        "install the optional value-phase work-counter buffer");
  m.def("set_value_ktables", &dblink::set_value_ktables,
        "install the k>=2 single-value perturbation tables");
  m.def("set_value_k2tables", &dblink::set_value_k2tables,
        "install the two-distinct-value (k, m) perturbation tables");
  m.def("mfma_score_bench", &dblink::mfma_score_bench,
        "MFMA one-hot categorical scorer (experiment)");
  m.def("scalar_score_bench", &dblink::scalar_score_bench,
        "LDS scalar categorical scorer (experiment)");
}
