// Python bindings for the MI355X HIP kernels (pure-ROCm torch extension —
// no hipify, no CUDA shims: this file uses torch's native c10::hip API and
// links the hipcc-compiled kernel objects).
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#define CHECK_DEV(x) TORCH_CHECK(x.is_cuda(), #x " must be on the GPU")
#define CHECK_CONTIG(x) TORCH_CHECK(x.is_contiguous(), #x " must be contiguous")

extern "C" {
void launch_hist_build(const void*, long, const int*, long, const float*,
                       const float*, float*, int, int, hipStream_t);
void launch_predict_forest(const int*, const float*, const int*, const int*,
                           const float*, const long*, const float*,
                           const float*, long, int, float*, int, int, int,
                           const int*, const unsigned*, hipStream_t);
void launch_predict_leaf(const int*, const float*, const int*, const int*,
                         const int*, const long*, const float*, long, int,
                         int*, int, const int*, const unsigned*, hipStream_t);
void launch_bin_matrix(const float*, const float*, long, int, int, int, void*,
                       hipStream_t);
void launch_hist_build_fixed(const void*, long, const int*, long, const float*,
                             const float*, long long*, int, int, double,
                             double, hipStream_t);
void launch_hist_build_fixed_pair(const void*, long, const int*, long,
                                  const float*, const float*, const void*,
                                  long long*, int, int, int, double, double,
                                  const int*, int, hipStream_t);
void launch_partition(const void*, long, const int*, long, int, int, int*,
                      int*, int*, hipStream_t);
void launch_split_scan(const float*, int, long, int, float, float, float,
                       float, float, long, const bool*, float*, float*,
                       hipStream_t);
void launch_vw_sgd(const int*, const float*, const long*, const float*,
                   const float*, float*, float*, float*, float, float, float,
                   int, int, long, float*, hipStream_t);
void launch_vw_predict(const int*, const float*, const long*, const float*,
                       long, float*, hipStream_t);
void launch_tree_shap(const int*, const float*, const int*, const int*,
                      const float*, const float*, const long*, const int*,
                      const unsigned*, const float*, long, int, int, int, int,
                      float*, hipStream_t);
void launch_csr_hist_fixed(const long*, const int*,
                           const unsigned char*, const long long*,
                           const long long*, const int*, long, long long*,
                           int, hipStream_t);
void launch_csr_gather_bin(const long*, const int*, const unsigned char*,
                           const int*, long, int, int, int*, hipStream_t);
void launch_csr_hist_fixed_v2(const long*, const int*, const unsigned char*,
                              const long long*, const long long*, const int*,
                              long, long long*, int, long long*, hipStream_t);
void launch_csr_hist_fixed_lds(const long*, const int*, const unsigned char*,
                               const long long*, const long long*,
                               const int*, long, long long*, int, int,
                               long long*, hipStream_t);
void launch_csr_partition(const long*, const int*, const unsigned char*,
                          const int*, long, int, int, int, const unsigned*,
                          unsigned char*, int*, int*, int*, hipStream_t);
}

static hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

torch::Tensor hist_build(torch::Tensor binned_i4, torch::Tensor rows,
                         torch::Tensor grad, torch::Tensor hess, long n_bins) {
  CHECK_DEV(binned_i4); CHECK_CONTIG(binned_i4);
  CHECK_DEV(rows); CHECK_CONTIG(rows);
  CHECK_DEV(grad); CHECK_CONTIG(grad);
  CHECK_DEV(hess); CHECK_CONTIG(hess);
  TORCH_CHECK(binned_i4.dtype() == torch::kUInt8, "binned must be uint8");
  TORCH_CHECK(rows.dtype() == torch::kInt32, "rows must be int32");
  const long ngroups = binned_i4.size(0);
  const long n_rows = binned_i4.size(1);
  auto hist = torch::zeros({ngroups * 4, n_bins, 3},
                           grad.options().dtype(torch::kFloat32));
  launch_hist_build(binned_i4.data_ptr(), n_rows, rows.data_ptr<int>(),
                    rows.numel(), grad.data_ptr<float>(),
                    hess.data_ptr<float>(), hist.data_ptr<float>(),
                    (int)n_bins, (int)ngroups, cur_stream());
  return hist;
}

torch::Tensor hist_build_fixed_pair(torch::Tensor binned_pair,
                                    torch::Tensor rows, torch::Tensor grad,
                                    torch::Tensor hess, long n_bins,
                                    long tail_bytes,
                                    double scale_g, double scale_h) {
  CHECK_DEV(binned_pair); CHECK_CONTIG(binned_pair);
  CHECK_DEV(rows); CHECK_CONTIG(rows);
  TORCH_CHECK(rows.dtype() == torch::kInt32, "rows must be int32");
  TORCH_CHECK(binned_pair.dtype() == torch::kInt64, "paired binned is int64");
  const long npairs = binned_pair.size(0);
  const long n_rows = binned_pair.size(1);
  auto hist = torch::zeros({npairs * 8, n_bins, 3},
                           grad.options().dtype(torch::kInt64));
  launch_hist_build_fixed_pair(binned_pair.data_ptr(), n_rows,
                               rows.data_ptr<int>(), rows.numel(),
                               grad.data_ptr<float>(), hess.data_ptr<float>(),
                               nullptr,
                               (long long*)hist.data_ptr<int64_t>(),
                               (int)n_bins, (int)npairs, (int)tail_bytes,
                               scale_g, scale_h, nullptr, -1, cur_stream());
  return hist;
}

torch::Tensor hist_build_fixed(torch::Tensor binned_i4, torch::Tensor rows,
                               torch::Tensor grad, torch::Tensor hess,
                               long n_bins, double scale_g, double scale_h) {
  CHECK_DEV(binned_i4); CHECK_CONTIG(binned_i4);
  CHECK_DEV(rows); CHECK_CONTIG(rows);
  TORCH_CHECK(rows.dtype() == torch::kInt32, "rows must be int32");
  const long ngroups = binned_i4.size(0);
  const long n_rows = binned_i4.size(1);
  auto hist = torch::zeros({ngroups * 4, n_bins, 3},
                           grad.options().dtype(torch::kInt64));
  launch_hist_build_fixed(binned_i4.data_ptr(), n_rows, rows.data_ptr<int>(),
                          rows.numel(), grad.data_ptr<float>(),
                          hess.data_ptr<float>(),
                          (long long*)hist.data_ptr<int64_t>(), (int)n_bins,
                          (int)ngroups, scale_g, scale_h, cur_stream());
  return hist;
}

static std::pair<const int*, const unsigned*> cat_ptrs(
    const c10::optional<torch::Tensor>& cat_offset,
    const c10::optional<torch::Tensor>& cat_words) {
  const int* co = nullptr;
  const unsigned* cw = nullptr;
  if (cat_offset.has_value() && cat_words.has_value()
      && cat_words->numel() > 0) {
    co = cat_offset->data_ptr<int>();
    cw = (const unsigned*)cat_words->data_ptr<int>();
  }
  return {co, cw};
}

torch::Tensor predict_forest(torch::Tensor feat, torch::Tensor thr,
                             torch::Tensor left, torch::Tensor right,
                             torch::Tensor val, torch::Tensor offsets,
                             torch::Tensor tw, torch::Tensor X,
                             long n_outputs, long t0, long t1,
                             c10::optional<torch::Tensor> cat_offset,
                             c10::optional<torch::Tensor> cat_words) {
  CHECK_DEV(X); CHECK_CONTIG(X);
  const long n = X.size(0);
  const long nf = X.size(1);
  auto out = torch::zeros({n, n_outputs}, X.options().dtype(torch::kFloat32));
  TORCH_CHECK(offsets.dtype() == torch::kInt64, "offsets must be int64");
  auto [co, cw] = cat_ptrs(cat_offset, cat_words);
  launch_predict_forest(feat.data_ptr<int>(), thr.data_ptr<float>(),
                        left.data_ptr<int>(), right.data_ptr<int>(),
                        val.data_ptr<float>(), offsets.data_ptr<long>(),
                        tw.data_ptr<float>(), X.data_ptr<float>(), n, (int)nf,
                        out.data_ptr<float>(), (int)n_outputs, (int)t0,
                        (int)t1, co, cw, cur_stream());
  return out;
}

torch::Tensor predict_leaf(torch::Tensor feat, torch::Tensor thr,
                           torch::Tensor left, torch::Tensor right,
                           torch::Tensor leaf_index, torch::Tensor offsets,
                           torch::Tensor X,
                           c10::optional<torch::Tensor> cat_offset,
                           c10::optional<torch::Tensor> cat_words) {
  CHECK_DEV(X); CHECK_CONTIG(X);
  const long n = X.size(0);
  const long nf = X.size(1);
  const long n_trees = offsets.numel() - 1;
  auto out = torch::zeros({n, n_trees}, X.options().dtype(torch::kInt32));
  auto [co, cw] = cat_ptrs(cat_offset, cat_words);
  launch_predict_leaf(feat.data_ptr<int>(), thr.data_ptr<float>(),
                      left.data_ptr<int>(), right.data_ptr<int>(),
                      leaf_index.data_ptr<int>(), offsets.data_ptr<long>(),
                      X.data_ptr<float>(), n, (int)nf, out.data_ptr<int>(),
                      (int)n_trees, co, cw, cur_stream());
  return out;
}

std::tuple<torch::Tensor, torch::Tensor> partition_rows(
    torch::Tensor binned_i4, torch::Tensor rows, long feature, long thr,
    long known_left) {
  CHECK_DEV(binned_i4); CHECK_CONTIG(binned_i4);
  CHECK_DEV(rows); CHECK_CONTIG(rows);
  TORCH_CHECK(rows.dtype() == torch::kInt32, "rows must be int32");
  const long m = rows.numel();
  auto out = torch::empty({m}, rows.options());
  auto scratch = torch::empty({4096}, rows.options());
  auto total = torch::zeros({1}, rows.options());
  if (m > 0) {
    launch_partition(binned_i4.data_ptr(), binned_i4.size(1),
                     rows.data_ptr<int>(), m, (int)feature, (int)thr,
                     out.data_ptr<int>(), scratch.data_ptr<int>(),
                     total.data_ptr<int>(), cur_stream());
  }
  // single-rank training already knows the exact left count from the split
  // stats (integer histogram counts) — skip the device→host sync entirely
  const long nl = known_left >= 0 ? known_left : total.item<int>();
  return {out.slice(0, 0, nl), out.slice(0, nl, m)};
}

torch::Tensor bin_matrix(torch::Tensor X, torch::Tensor ub, long n_bins) {
  CHECK_DEV(X); CHECK_CONTIG(X);
  CHECK_DEV(ub); CHECK_CONTIG(ub);
  const long n = X.size(0);
  const long nf = X.size(1);
  const long ngroups = (nf + 3) / 4;
  auto out = torch::zeros({ngroups, n, 4}, X.options().dtype(torch::kUInt8));
  launch_bin_matrix(X.data_ptr<float>(), ub.data_ptr<float>(), n, (int)nf,
                    (int)n_bins, (int)ngroups, out.data_ptr(), cur_stream());
  return out;
}

torch::Tensor split_scan(torch::Tensor hists, long n_bins, double l1,
                         double l2, double min_data, double min_hess,
                         double min_gain, long nf_real,
                         c10::optional<torch::Tensor> feat_mask) {
  // hists: (n_hists, nf_pad, n_bins, 3) — returns (n_hists, 6) on device:
  // {gain, feature, bin, GL, HL, CL}
  CHECK_DEV(hists); CHECK_CONTIG(hists);
  const long n_hists = hists.size(0);
  const long nf_pad = hists.size(1);
  auto scratch = torch::empty({n_hists, nf_pad, 6}, hists.options());
  auto out = torch::empty({n_hists, 6}, hists.options());
  const bool* mask_ptr = nullptr;
  if (feat_mask.has_value()) {
    TORCH_CHECK(feat_mask->dtype() == torch::kBool, "feat_mask must be bool");
    mask_ptr = feat_mask->data_ptr<bool>();
  }
  launch_split_scan(hists.data_ptr<float>(), (int)n_hists, nf_pad,
                    (int)n_bins, (float)l1, (float)l2, (float)min_data,
                    (float)min_hess, (float)min_gain, nf_real, mask_ptr,
                    scratch.data_ptr<float>(), out.data_ptr<float>(),
                    cur_stream());
  return out;
}

torch::Tensor vw_sgd_minibatch(torch::Tensor idx, torch::Tensor val,
                               torch::Tensor off, torch::Tensor label,
                               torch::Tensor w_tbl, torch::Tensor g_tbl,
                               double lr, double l2, double power_t,
                               long loss,
                               c10::optional<torch::Tensor> ex_weight,
                               c10::optional<torch::Tensor> s_tbl,
                               bool invariant) {
  CHECK_DEV(w_tbl); CHECK_CONTIG(w_tbl);
  const long n_ex = off.numel() - 1;
  auto preds = torch::zeros({n_ex}, w_tbl.options());
  const float* wptr = ex_weight.has_value() ? ex_weight->data_ptr<float>()
                                            : nullptr;
  float* sptr = s_tbl.has_value() ? s_tbl->data_ptr<float>() : nullptr;
  launch_vw_sgd(idx.data_ptr<int>(), val.data_ptr<float>(),
                off.data_ptr<long>(), label.data_ptr<float>(), wptr,
                w_tbl.data_ptr<float>(), g_tbl.data_ptr<float>(), sptr,
                (float)lr, (float)l2, (float)power_t, (int)loss,
                invariant ? 1 : 0, n_ex,
                preds.data_ptr<float>(), cur_stream());
  return preds;
}

torch::Tensor vw_predict(torch::Tensor idx, torch::Tensor val,
                         torch::Tensor off, torch::Tensor w_tbl) {
  CHECK_DEV(w_tbl); CHECK_CONTIG(w_tbl);
  const long n_ex = off.numel() - 1;
  auto out = torch::zeros({n_ex}, w_tbl.options());
  launch_vw_predict(idx.data_ptr<int>(), val.data_ptr<float>(),
                    off.data_ptr<long>(), w_tbl.data_ptr<float>(), n_ex,
                    out.data_ptr<float>(), cur_stream());
  return out;
}

torch::Tensor tree_shap(torch::Tensor feat, torch::Tensor thr,
                        torch::Tensor left, torch::Tensor right,
                        torch::Tensor val, torch::Tensor cnt,
                        torch::Tensor offsets, torch::Tensor X,
                        long n_outputs, long max_depth,
                        c10::optional<torch::Tensor> cat_offset,
                        c10::optional<torch::Tensor> cat_words) {
  CHECK_DEV(X); CHECK_CONTIG(X);
  const long n = X.size(0);
  const long nf = X.size(1);
  const long n_trees = offsets.numel() - 1;
  auto out = torch::zeros({n, n_outputs * (nf + 1)},
                          X.options().dtype(torch::kFloat32));
  auto [co, cw] = cat_ptrs(cat_offset, cat_words);
  launch_tree_shap(feat.data_ptr<int>(), thr.data_ptr<float>(),
                   left.data_ptr<int>(), right.data_ptr<int>(),
                   val.data_ptr<float>(), cnt.data_ptr<float>(),
                   offsets.data_ptr<long>(), co, cw, X.data_ptr<float>(), n,
                   (int)nf, (int)n_trees, (int)n_outputs, (int)max_depth,
                   out.data_ptr<float>(), cur_stream());
  return out;
}

// --------------------------------------------------------------- sparse CSR
torch::Tensor csr_hist_fixed(torch::Tensor indptr, torch::Tensor col,
                             torch::Tensor binv, torch::Tensor gq,
                             torch::Tensor hq, torch::Tensor rows,
                             long nf, long n_bins) {
  CHECK_DEV(indptr); CHECK_CONTIG(indptr);
  CHECK_DEV(col); CHECK_CONTIG(col);
  CHECK_DEV(binv); CHECK_CONTIG(binv);
  CHECK_DEV(gq); CHECK_CONTIG(gq);
  CHECK_DEV(hq); CHECK_CONTIG(hq);
  CHECK_DEV(rows); CHECK_CONTIG(rows);
  TORCH_CHECK(indptr.dtype() == torch::kInt64, "indptr must be int64");
  TORCH_CHECK(col.dtype() == torch::kInt32, "col must be int32");
  TORCH_CHECK(binv.dtype() == torch::kUInt8, "binv must be uint8");
  TORCH_CHECK(gq.dtype() == torch::kInt64 && hq.dtype() == torch::kInt64,
              "gq/hq must be int64 fixed point");
  auto hist = torch::zeros({nf, n_bins, 3},
                           gq.options().dtype(torch::kInt64));
  launch_csr_hist_fixed(indptr.data_ptr<long>(), col.data_ptr<int>(),
                        binv.data_ptr<unsigned char>(),
                        (const long long*)gq.data_ptr<int64_t>(),
                        (const long long*)hq.data_ptr<int64_t>(),
                        rows.data_ptr<int>(), rows.numel(),
                        (long long*)hist.data_ptr<int64_t>(), (int)n_bins,
                        cur_stream());
  return hist;
}

// v2: wave-cooperative row chunks + in-kernel leaf totals
std::tuple<torch::Tensor, torch::Tensor> csr_hist_fixed_tot(
    torch::Tensor indptr, torch::Tensor col, torch::Tensor binv,
    torch::Tensor gq, torch::Tensor hq, torch::Tensor rows, long nf,
    long n_bins) {
  CHECK_DEV(indptr); CHECK_CONTIG(indptr);
  CHECK_DEV(col); CHECK_CONTIG(col);
  CHECK_DEV(binv); CHECK_CONTIG(binv);
  CHECK_DEV(gq); CHECK_CONTIG(gq);
  CHECK_DEV(hq); CHECK_CONTIG(hq);
  CHECK_DEV(rows); CHECK_CONTIG(rows);
  auto hist = torch::zeros({nf, n_bins, 3},
                           gq.options().dtype(torch::kInt64));
  auto tot = torch::zeros({3}, gq.options().dtype(torch::kInt64));
  if (nf <= 2048) {
    // small/medium nf: LDS-privatized feature chunks (redundant coalesced
    // reads, 9x-faster LDS atomics, one global flush per cell)
    launch_csr_hist_fixed_lds(indptr.data_ptr<long>(), col.data_ptr<int>(),
                              binv.data_ptr<unsigned char>(),
                              (const long long*)gq.data_ptr<int64_t>(),
                              (const long long*)hq.data_ptr<int64_t>(),
                              rows.data_ptr<int>(), rows.numel(),
                              (long long*)hist.data_ptr<int64_t>(),
                              (int)n_bins, (int)nf,
                              (long long*)tot.data_ptr<int64_t>(),
                              cur_stream());
  } else {
    // wide-nf: atomics spread across nf*n_bins cells — contention is low
    launch_csr_hist_fixed_v2(indptr.data_ptr<long>(), col.data_ptr<int>(),
                             binv.data_ptr<unsigned char>(),
                             (const long long*)gq.data_ptr<int64_t>(),
                             (const long long*)hq.data_ptr<int64_t>(),
                             rows.data_ptr<int>(), rows.numel(),
                             (long long*)hist.data_ptr<int64_t>(), (int)n_bins,
                             (long long*)tot.data_ptr<int64_t>(),
                             cur_stream());
  }
  return {hist, tot};
}

torch::Tensor csr_gather_bins(torch::Tensor indptr, torch::Tensor col,
                              torch::Tensor binv, torch::Tensor rows,
                              long feature, long zero_bin) {
  CHECK_DEV(indptr); CHECK_CONTIG(indptr);
  CHECK_DEV(col); CHECK_CONTIG(col);
  CHECK_DEV(binv); CHECK_CONTIG(binv);
  CHECK_DEV(rows); CHECK_CONTIG(rows);
  auto out = torch::empty({rows.numel()},
                          rows.options().dtype(torch::kInt32));
  launch_csr_gather_bin(indptr.data_ptr<long>(), col.data_ptr<int>(),
                        binv.data_ptr<unsigned char>(), rows.data_ptr<int>(),
                        rows.numel(), (int)feature, (int)zero_bin,
                        out.data_ptr<int>(), cur_stream());
  return out;
}

std::tuple<torch::Tensor, torch::Tensor> csr_partition_rows(
    torch::Tensor indptr, torch::Tensor col, torch::Tensor binv,
    torch::Tensor rows, long feature, long zero_bin, long thr,
    long known_left) {
  CHECK_DEV(indptr); CHECK_CONTIG(indptr);
  CHECK_DEV(col); CHECK_CONTIG(col);
  CHECK_DEV(binv); CHECK_CONTIG(binv);
  CHECK_DEV(rows); CHECK_CONTIG(rows);
  const long m = rows.numel();
  auto out = torch::empty({m}, rows.options());
  auto pred = torch::empty({m}, rows.options().dtype(torch::kUInt8));
  auto scratch = torch::empty({4096}, rows.options());
  auto total = torch::zeros({1}, rows.options());
  if (m > 0) {
    launch_csr_partition(indptr.data_ptr<long>(), col.data_ptr<int>(),
                         binv.data_ptr<unsigned char>(), rows.data_ptr<int>(),
                         m, (int)feature, (int)zero_bin, (int)thr, nullptr,
                         pred.data_ptr<unsigned char>(),
                         out.data_ptr<int>(), scratch.data_ptr<int>(),
                         total.data_ptr<int>(), cur_stream());
  }
  const long nl = known_left >= 0 ? known_left : total.item<int>();
  return {out.slice(0, 0, nl), out.slice(0, nl, m)};
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("hist_build", &hist_build, "per-leaf (feature,bin) grad/hess/count histogram");
  m.def("hist_build_fixed_pair", &hist_build_fixed_pair,
        "fixed-point histogram over paired (8-feature) planes");
  m.def("hist_build_fixed", &hist_build_fixed,
        "fixed-point u64 histogram (fast LDS integer atomics)");
  m.def("predict_forest", &predict_forest, "GBDT ensemble raw scores");
  m.def("predict_leaf", &predict_leaf, "GBDT per-tree leaf indices");
  m.def("bin_matrix", &bin_matrix, "quantile binning to interleaved uint8");
  m.def("split_scan", &split_scan, "fused best-split over sibling histograms");
  m.def("partition_rows", &partition_rows,
        "stable ordered row partition, single sync");
  m.def("vw_sgd_minibatch", &vw_sgd_minibatch, "adaptive sparse SGD minibatch");
  m.def("vw_predict", &vw_predict, "sparse linear predict");
  m.def("tree_shap", &tree_shap, "path-dependent TreeSHAP contributions");
  m.def("csr_hist_fixed", &csr_hist_fixed,
        "fixed-point histogram over stored CSR entries");
  m.def("csr_gather_bins", &csr_gather_bins,
        "per-row bin of one feature from CSR (missing -> zero bin)");
  m.def("csr_partition_rows", &csr_partition_rows,
        "fused CSR leaf partition (binary-search predicate, single sync)");
  m.def("csr_hist_fixed_tot", &csr_hist_fixed_tot,
        "wave-cooperative CSR histogram + in-kernel leaf totals");
}
