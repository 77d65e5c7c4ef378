// Native leaf-wise tree-growth driver — the C++ runtime piece of the GBDT.
//
// Runs the entire leaf-wise loop in C++ with ONE Python call per tree,
// organized around two ideas:
//
//  * ARENA MODE: the tree's working set (paired bin planes + pre-quantized
//    (gq,hq) records + original row ids) lives in a per-tree DOUBLE-BUFFERED
//    arena.  Every split scatters the parent's segment into the other buffer
//    (stable left|right), so child histograms read CONTIGUOUS rows — the
//    sparse row gather that bounded child hist builds (measured 2.3x,
//    tools/hist_gather_probe.py) is gone; the cost is streaming ~124 B/row
//    of arena payload per split, which HBM3E absorbs.  Live leaves always
//    own disjoint row ranges, so two in-flight splits never overlap even
//    across buffers.
//
//  * SPECULATION: a popped leaf's split work is independent of processing
//    order (its split was fixed at creation), so the driver pre-launches the
//    next-best candidate's whole chain (partition → smaller-child hist →
//    sibling subtraction → scan → pinned readback behind a HIP event) while
//    the host waits on the current readback.  Commit order stays exactly
//    leaf-wise.  The chain never blocks the host: child ranges resolve from
//    the DEVICE-side left count, which rides back with the scan result.
//    Candidate order is deterministic across ranks, so the histogram
//    all_reduce order matches and speculation is safe in multi-rank mode.
//
// Categorical features fall back to the Python grower (models/gbdt/
// trainer.py).  Replaces the growth loop the reference delegates to
// LightGBM's serial_tree_learner behind LGBM_BoosterUpdateOneIter
// (SURVEY §2.1).
#include <torch/extension.h>
#include <torch/csrc/distributed/c10d/ProcessGroup.hpp>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#include <algorithm>
#include <cmath>
#include <memory>
#include <numeric>
#include <queue>
#include <vector>

extern "C" {
void launch_arena_gather(const void*, long, const float*, const float*,
                         const int*, long, int, double, double, void*, void*,
                         int*, hipStream_t);
void launch_partition_arena(const void*, const void*, const int*, void*,
                            void*, int*, int*, long, int, long, long, int,
                            int, const unsigned*, int*, int*, hipStream_t);
void launch_hist_pair_range(const void*, const void*, long, long, long,
                            long long*, int, int, int, const int*, int,
                            hipStream_t);
void launch_split_scan_fixed(const long long*, int, long, int, float, float,
                             float, float, float, long, const bool*, float*,
                             float*, double, double, hipStream_t);
}

static hipStream_t grower_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

namespace {

struct GrowCtx {
  long n_arena;    // rows in the arena (= rows_root size after bagging)
  int n_bins;
  int npairs;
  int tail_bytes;  // valid feature-bytes in the last pair (zero-pad skipped)
  // double-buffered arena: [0]/[1]
  torch::Tensor pair[2];   // (npairs, n_arena) i64 paired bin planes
  torch::Tensor ghq[2];    // (n_arena, 2) i64 (gq, CNT|hq)
  torch::Tensor rowid[2];  // (n_arena,) i32 original row ids
  long nf;
  double scale_g, scale_h;
  double l1, l2, min_data, min_hess, min_gain;
  int num_leaves, max_depth;
  torch::Tensor feat_mask;  // bool (nf_pad,) or undefined
  // c10d process group for the per-split histogram all_reduce — called from
  // C++ with NO GIL (the round-1 Python-callback hop is gone).  With the
  // NCCL(=RCCL) backend Work::wait() only blocks the current HIP stream, so
  // speculated histogram kernels keep overlapping the collective.
  c10::intrusive_ptr<c10d::ProcessGroup> pg;
  bool has_reduce;
  bool distributed;
  // categorical features (already filtered by the iteration's feature mask);
  // the host-side sorted one-vs-rest scan replicates trainer._cat_scan
  // bit-for-bit (float32 arithmetic, lexsort tie-break by bin id)
  std::vector<int> cat_feats;
  double cat_smooth = 10.0;
  torch::Tensor cat_idx_dev;  // i64 device indices for index_select
  torch::Tensor scratch;    // partition block counters (stream-ordered reuse)
  torch::Tensor total;      // partition left-count (stream-ordered reuse)
  torch::Tensor dst_idx;    // (n_arena,) i32 per-split dest ranks (reused)
};

struct SplitJob {
  torch::Tensor hist_l, hist_r;
  torch::Tensor scan_host;  // (2,6) f32 pinned
  torch::Tensor nl_host;    // (1,) i32 pinned — local left count readback
  torch::Tensor cat_host;   // (nh, ncat, nb, 3) i64 pinned — cat hist rows
  torch::Tensor bits_host;  // (8,) i32 pinned — partition bitset staging
  torch::Tensor bits_dev;   // (8,) i32 device
  long nl_known = -1;       // host-known left count (single-rank fast path)
  hipEvent_t ev = nullptr;

  ~SplitJob() {
    if (ev) (void)hipEventDestroy(ev);
  }
};

struct LeafCand {
  double gain = 0;
  long seq = 0;
  int node_id = 0;
  int depth = 0;
  long lo = 0, hi = 0;  // arena row range
  int buf = 0;          // which arena buffer holds this leaf's rows
  torch::Tensor hist;   // int64 (nf_pad, nb, 3), globally reduced
  double G = 0, H = 0, C = 0;
  double GL = 0, HL = 0, CL = 0;
  int feat = -1, bin = 0;
  std::vector<int> cats;  // chosen category bins (empty = numeric split)
  std::shared_ptr<SplitJob> job;
};

using CandPtr = std::shared_ptr<LeafCand>;

struct CandCmp {
  bool operator()(const CandPtr& a, const CandPtr& b) const {
    if (a->gain != b->gain) return a->gain < b->gain;  // max-heap by gain
    return a->seq > b->seq;
  }
};

// histogram over an arena segment [lo+?, ...) of buffer `buf`; side >= 0
// resolves the child sub-range from the device left count (no host sync)
torch::Tensor build_hist(GrowCtx& ctx, int buf, long lo, long m,
                         const int* nl_dev = nullptr, int side = -1) {
  auto hist = torch::zeros({ctx.npairs * 8, ctx.n_bins, 3},
                           ctx.pair[buf].options());
  launch_hist_pair_range(
      ctx.pair[buf].data_ptr(), ctx.ghq[buf].data_ptr(), ctx.n_arena, lo, m,
      (long long*)hist.data_ptr<int64_t>(), ctx.n_bins, ctx.npairs,
      ctx.tail_bytes, nl_dev, side, grower_stream());
  if (ctx.has_reduce) {
    std::vector<at::Tensor> v{hist};
    ctx.pg->allreduce(v)->wait();  // stream-ordered on RCCL; no GIL
  }
  return hist;
}

// launch the (2, nf_pad) scan of stacked child histograms; result lands in a
// pinned host buffer behind an event — no stream sync
void launch_scan_async(GrowCtx& ctx, const torch::Tensor& hists_i64,
                       SplitJob& job) {
  const long nh = hists_i64.size(0);
  const long nf_pad = hists_i64.size(1);
  auto fopt = torch::TensorOptions()
                  .dtype(torch::kFloat32).device(hists_i64.device());
  auto scratch = torch::empty({nh, nf_pad, 6}, fopt);
  auto out = torch::empty({nh, 6}, fopt);
  const bool* mask = ctx.feat_mask.defined()
                         ? ctx.feat_mask.data_ptr<bool>() : nullptr;
  launch_split_scan_fixed(
      (const long long*)hists_i64.data_ptr<int64_t>(), (int)nh, nf_pad,
      ctx.n_bins, (float)ctx.l1, (float)ctx.l2, (float)ctx.min_data,
      (float)ctx.min_hess, (float)ctx.min_gain, ctx.nf, mask,
      scratch.data_ptr<float>(), out.data_ptr<float>(), 1.0 / ctx.scale_g,
      1.0 / ctx.scale_h, grower_stream());
  job.scan_host = torch::empty({nh, 6}, torch::TensorOptions()
                                            .dtype(torch::kFloat32)
                                            .pinned_memory(true));
  (void)hipMemcpyAsync(job.scan_host.data_ptr<float>(), out.data_ptr<float>(),
                       nh * 6 * sizeof(float), hipMemcpyDeviceToHost,
                       grower_stream());
  if (!ctx.cat_feats.empty()) {
    // categorical rows of the reduced histogram ride back with the scan;
    // the sorted one-vs-rest scan runs on host at commit (≤ ncat×256 bins)
    const long ncat = (long)ctx.cat_feats.size();
    auto cat_rows = hists_i64.index_select(1, ctx.cat_idx_dev).contiguous();
    job.cat_host = torch::empty({nh, ncat, (long)ctx.n_bins, 3},
                                torch::TensorOptions()
                                    .dtype(torch::kInt64)
                                    .pinned_memory(true));
    (void)hipMemcpyAsync(job.cat_host.data_ptr<int64_t>(),
                         cat_rows.data_ptr<int64_t>(),
                         nh * ncat * ctx.n_bins * 3 * sizeof(int64_t),
                         hipMemcpyDeviceToHost, grower_stream());
  }
  (void)hipEventCreateWithFlags(&job.ev, hipEventDisableTiming);
  (void)hipEventRecord(job.ev, grower_stream());
}

struct CatBest {
  float gain = -INFINITY;
  int feat = -1;
  float GL = 0, HL = 0, CL = 0;
  std::vector<int> cats;
};

// Bit-for-bit replica of trainer._cat_scan: float32 arithmetic, sequential
// float32 cumsums, lexsort tie-break by bin id, first-max argmax.
CatBest cat_scan_host(const GrowCtx& ctx, const int64_t* hist) {
  CatBest best;
  const int nb = ctx.n_bins;
  const double inv_g = 1.0 / ctx.scale_g, inv_h = 1.0 / ctx.scale_h;
  for (size_t ci = 0; ci < ctx.cat_feats.size(); ++ci) {
    const int64_t* h = hist + ci * (size_t)nb * 3;
    std::vector<int> present;
    for (int b = 0; b < nb; ++b)
      if (h[b * 3 + 2] > 0) present.push_back(b);
    const int m = (int)present.size();
    if (m < 2) continue;
    std::vector<float> g(m), hh(m), c(m), ratio(m);
    for (int i = 0; i < m; ++i) {
      const int b = present[i];
      g[i] = (float)((double)h[b * 3 + 0] * inv_g);
      hh[i] = (float)((double)h[b * 3 + 1] * inv_h);
      c[i] = (float)h[b * 3 + 2];
      ratio[i] = g[i] / (hh[i] + (float)ctx.cat_smooth);
    }
    std::vector<int> order(m);
    std::iota(order.begin(), order.end(), 0);
    std::stable_sort(order.begin(), order.end(), [&](int x, int y) {
      if (ratio[x] != ratio[y]) return ratio[x] < ratio[y];
      return present[x] < present[y];
    });
    std::vector<float> GL(m), HL(m), CL(m);
    float ag = 0.f, ah = 0.f, ac = 0.f;
    for (int i = 0; i < m; ++i) {
      ag += g[order[i]]; ah += hh[order[i]]; ac += c[order[i]];
      GL[i] = ag; HL[i] = ah; CL[i] = ac;
    }
    const float G = GL[m - 1], H = HL[m - 1], C = CL[m - 1];
    auto sc = [&](float Gs, float Hs) {
      const float Ga = std::max(std::fabs(Gs) - (float)ctx.l1, 0.0f);
      return Ga * Ga / (Hs + (float)ctx.l2 + 1e-32f);
    };
    const float scGH = sc(G, H);
    int bi = -1;
    float bg = -INFINITY;
    for (int i = 0; i < m - 1; ++i) {
      const bool valid = CL[i] >= (float)ctx.min_data
          && (C - CL[i]) >= (float)ctx.min_data
          && HL[i] >= (float)ctx.min_hess
          && (H - HL[i]) >= (float)ctx.min_hess;
      const float gain = valid
          ? sc(GL[i], HL[i]) + sc(G - GL[i], H - HL[i]) - scGH
          : -INFINITY;
      if (gain > bg) { bg = gain; bi = i; }
    }
    if (bi < 0 || !std::isfinite(bg)) continue;
    if (best.feat < 0 || bg > best.gain) {
      best.gain = bg;
      best.feat = ctx.cat_feats[ci];
      best.GL = GL[bi]; best.HL = HL[bi]; best.CL = CL[bi];
      best.cats.assign(order.begin(), order.begin() + bi + 1);
      for (auto& v : best.cats) v = present[v];
    }
  }
  return best;
}

// Fill a candidate's split from the readbacks: numeric scan row `hi`, then
// let the categorical best take over iff strictly better (trainer._scan).
void resolve_best(const GrowCtx& ctx, SplitJob& job, int hi, LeafCand& c) {
  auto a = job.scan_host.accessor<float, 2>();
  c.gain = a[hi][0]; c.feat = (int)a[hi][1]; c.bin = (int)a[hi][2];
  c.GL = a[hi][3]; c.HL = a[hi][4]; c.CL = a[hi][5];
  c.cats.clear();
  if (!ctx.cat_feats.empty()) {
    const size_t stride =
        ctx.cat_feats.size() * (size_t)ctx.n_bins * 3;
    CatBest cb = cat_scan_host(
        ctx, job.cat_host.data_ptr<int64_t>() + (size_t)hi * stride);
    if (cb.feat >= 0 && (double)cb.gain > c.gain) {
      c.gain = (double)cb.gain; c.feat = cb.feat; c.bin = 0;
      c.GL = (double)cb.GL; c.HL = (double)cb.HL; c.CL = (double)cb.CL;
      c.cats = std::move(cb.cats);
    }
  }
}

void launch_job(GrowCtx& ctx, LeafCand& leaf) {
  auto job = std::make_shared<SplitJob>();
  const long m = leaf.hi - leaf.lo;
  const int src = leaf.buf, dst = leaf.buf ^ 1;
  const unsigned* cat_bits = nullptr;
  if (!leaf.cats.empty()) {
    // categorical split: stage the 256-bit category set and partition by
    // bitset test instead of bin <= thr
    job->bits_host = torch::zeros({8}, torch::TensorOptions()
                                           .dtype(torch::kInt32)
                                           .pinned_memory(true));
    unsigned* wb = (unsigned*)job->bits_host.data_ptr<int>();
    for (int b : leaf.cats) wb[b >> 5] |= (1u << (b & 31));
    job->bits_dev = torch::empty({8}, ctx.rowid[0].options()
                                          .dtype(torch::kInt32));
    (void)hipMemcpyAsync(job->bits_dev.data_ptr<int>(), wb,
                         8 * sizeof(int), hipMemcpyHostToDevice,
                         grower_stream());
    cat_bits = (const unsigned*)job->bits_dev.data_ptr<int>();
  }
  launch_partition_arena(
      ctx.pair[src].data_ptr(), ctx.ghq[src].data_ptr(),
      ctx.rowid[src].data_ptr<int>(), ctx.pair[dst].data_ptr(),
      ctx.ghq[dst].data_ptr(), ctx.rowid[dst].data_ptr<int>(),
      ctx.dst_idx.data_ptr<int>() + leaf.lo, ctx.n_arena,
      ctx.npairs, leaf.lo, m, leaf.feat, leaf.bin, cat_bits,
      ctx.scratch.data_ptr<int>(), ctx.total.data_ptr<int>(),
      grower_stream());

  const double CL = leaf.CL, CR = leaf.C - leaf.CL;
  const bool left_small = CL <= CR;  // by GLOBAL counts: same on all ranks
  torch::Tensor hist_small;
  if (!ctx.distributed && leaf.C < 1.6e7) {
    job->nl_known = (long)leaf.CL;  // exact integer counts on this rank
    const long lo_s = left_small ? leaf.lo : leaf.lo + job->nl_known;
    const long m_s = left_small ? job->nl_known : m - job->nl_known;
    hist_small = build_hist(ctx, dst, lo_s, m_s);
  } else {
    hist_small = build_hist(ctx, dst, leaf.lo, m,
                            ctx.total.data_ptr<int>(), left_small ? 0 : 1);
    job->nl_host = torch::empty({1}, torch::TensorOptions()
                                         .dtype(torch::kInt32)
                                         .pinned_memory(true));
    (void)hipMemcpyAsync(job->nl_host.data_ptr<int>(),
                         ctx.total.data_ptr<int>(), sizeof(int),
                         hipMemcpyDeviceToHost, grower_stream());
  }
  auto hist_big = leaf.hist - hist_small;
  job->hist_l = left_small ? hist_small : hist_big;
  job->hist_r = left_small ? hist_big : hist_small;
  launch_scan_async(ctx, torch::stack({job->hist_l, job->hist_r}), *job);
  leaf.job = job;
}

double leaf_output(double G, double H, double l1, double l2,
                   double max_delta) {
  double g = std::abs(G) - l1;
  if (g <= 0) return 0.0;
  double denom = H + l2;
  if (denom <= 0) return 0.0;  // all-zero quantized hessians, no L2
  double w = -std::copysign(g, G) / denom;
  if (max_delta > 0) w = std::clamp(w, -max_delta, max_delta);
  return w;
}

bool splittable(const GrowCtx& ctx, const LeafCand& c) {
  if (!(c.gain > ctx.min_gain) || !std::isfinite(c.gain)) return false;
  if (ctx.max_depth > 0 && c.depth >= ctx.max_depth) return false;
  return true;
}

}  // namespace

// Returns dict with node arrays (CPU int32/f32 tensors), per-leaf rows
// (device int32, concatenated) + offsets + leaf node ids.
py::dict grow_tree_native(torch::Tensor binned, torch::Tensor binned_pair,
                          torch::Tensor rows_root,
                          torch::Tensor grad, torch::Tensor hess,
                          long n_bins, long nf, double scale_g, double scale_h,
                          double l1, double l2, double min_data,
                          double min_hess, double min_gain, double max_delta,
                          long num_leaves, long max_depth,
                          c10::optional<torch::Tensor> feat_mask,
                          c10::optional<torch::Tensor> cat_feats,
                          double cat_smooth,
                          py::object process_group, bool distributed) {
  GrowCtx ctx;
  const long n_full = binned_pair.size(1);
  ctx.n_arena = rows_root.numel();
  ctx.n_bins = (int)n_bins;
  ctx.npairs = (int)binned_pair.size(0);
  ctx.tail_bytes = (int)(binned.size(0) * 4 - (ctx.npairs - 1) * 8);
  ctx.nf = nf;
  ctx.scale_g = scale_g;
  ctx.scale_h = scale_h;
  ctx.l1 = l1; ctx.l2 = l2;
  ctx.min_data = min_data; ctx.min_hess = min_hess; ctx.min_gain = min_gain;
  ctx.num_leaves = (int)num_leaves;
  ctx.max_depth = (int)max_depth;
  if (feat_mask.has_value()) ctx.feat_mask = *feat_mask;
  if (cat_feats.has_value() && cat_feats->numel() > 0) {
    auto cf = cat_feats->to(torch::kInt64).contiguous().cpu();
    auto acc = cf.accessor<int64_t, 1>();
    for (long i = 0; i < cf.numel(); ++i)
      ctx.cat_feats.push_back((int)acc[i]);
    ctx.cat_idx_dev = cf.to(binned.device());
    ctx.cat_smooth = cat_smooth;
  }
  if (!process_group.is_none()) {
    ctx.pg = process_group.cast<c10::intrusive_ptr<c10d::ProcessGroup>>();
  }
  ctx.has_reduce = ctx.pg && ctx.pg->getSize() > 1;
  ctx.distributed = distributed;

  py::gil_scoped_release nogil;

  ctx.scratch = torch::empty({4096}, rows_root.options().dtype(torch::kInt32));
  ctx.total = torch::zeros({1}, rows_root.options().dtype(torch::kInt32));
  // per-leaf segments are disjoint, so one dst_idx array serves concurrent
  // in-flight splits (indexed at leaf.lo)
  ctx.dst_idx = torch::empty({ctx.n_arena},
                             rows_root.options().dtype(torch::kInt32));
  auto i64d = rows_root.options().dtype(torch::kInt64);
  for (int b = 0; b < 2; ++b) {
    ctx.pair[b] = torch::empty({ctx.npairs, ctx.n_arena}, i64d);
    ctx.ghq[b] = torch::empty({ctx.n_arena, 2}, i64d);
    ctx.rowid[b] = torch::empty({ctx.n_arena},
                                rows_root.options().dtype(torch::kInt32));
  }
  // materialize arena buffer 0: bin pairs + quantized (gq,hq) + row ids for
  // the (possibly bagged) root row set
  launch_arena_gather(binned_pair.data_ptr(), n_full,
                      grad.data_ptr<float>(), hess.data_ptr<float>(),
                      rows_root.data_ptr<int>(), ctx.n_arena, ctx.npairs,
                      ctx.scale_g, ctx.scale_h, ctx.pair[0].data_ptr(),
                      ctx.ghq[0].data_ptr(), ctx.rowid[0].data_ptr<int>(),
                      grower_stream());

  std::vector<int> feature_, thr_bin_, left_, right_, leaf_idx_;
  std::vector<int> cat_off_, cat_words_;
  std::vector<float> value_, count_, gain_;
  auto new_node = [&]() {
    feature_.push_back(-1); thr_bin_.push_back(0); left_.push_back(-1);
    right_.push_back(-1); value_.push_back(0.f); count_.push_back(0.f);
    gain_.push_back(0.f); leaf_idx_.push_back(-1); cat_off_.push_back(-1);
    return (int)feature_.size() - 1;
  };

  auto root_hist = build_hist(ctx, 0, 0, ctx.n_arena);
  auto sums = root_hist.select(0, 0).sum(0).to(torch::kCPU);
  auto sa = sums.accessor<int64_t, 1>();
  const double G0 = (double)sa[0] / scale_g;
  const double H0 = (double)sa[1] / scale_h;
  const double C0 = (double)sa[2];

  auto root = std::make_shared<LeafCand>();
  root->node_id = new_node();
  root->lo = 0; root->hi = ctx.n_arena; root->buf = 0;
  root->hist = root_hist;
  root->G = G0; root->H = H0; root->C = C0;
  {
    // one-time root scan through the same async machinery
    SplitJob j;
    launch_scan_async(ctx, root_hist.unsqueeze(0), j);
    (void)hipEventSynchronize(j.ev);
    resolve_best(ctx, j, 0, *root);
  }
  count_[root->node_id] = (float)C0;
  value_[root->node_id] = (float)leaf_output(G0, H0, l1, l2, max_delta);

  std::priority_queue<CandPtr, std::vector<CandPtr>, CandCmp> heap;
  std::vector<CandPtr> finals;
  heap.push(root);
  finals.push_back(root);
  long seq = 1;
  int n_leaves = 1;

  while (n_leaves < ctx.num_leaves && !heap.empty()) {
    CandPtr leaf = heap.top();
    heap.pop();
    if (!splittable(ctx, *leaf)) continue;
    for (size_t i = 0; i < finals.size(); ++i)
      if (finals[i]->node_id == leaf->node_id) {
        finals.erase(finals.begin() + i);
        break;
      }

    if (!leaf->job) launch_job(ctx, *leaf);
    // speculate: pre-launch the next-best candidate's chain so its GPU work
    // overlaps this readback (exact commit order preserved; the cache is
    // consumed whenever that leaf is popped; live leaves own disjoint row
    // ranges so in-flight scatters never collide)
    if (n_leaves + 1 < ctx.num_leaves && !heap.empty()) {
      CandPtr nxt = heap.top();
      if (!nxt->job && splittable(ctx, *nxt)) launch_job(ctx, *nxt);
      if (n_leaves + 2 < ctx.num_leaves && heap.size() >= 2) {
        heap.pop();  // peek the second-best, then restore
        CandPtr nxt2 = heap.top();
        heap.push(nxt);
        if (!nxt2->job && splittable(ctx, *nxt2)) launch_job(ctx, *nxt2);
      }
    }

    SplitJob& job = *leaf->job;
    (void)hipEventSynchronize(job.ev);
    const long m_parent = leaf->hi - leaf->lo;
    const long nl = job.nl_known >= 0
                        ? job.nl_known
                        : (long)job.nl_host.data_ptr<int>()[0];

    const double GL = leaf->GL, HL = leaf->HL, CL = leaf->CL;
    const double GR = leaf->G - GL, HR = leaf->H - HL, CR = leaf->C - CL;
    const int nid = leaf->node_id;
    feature_[nid] = leaf->feat;
    thr_bin_[nid] = leaf->bin;
    gain_[nid] = (float)leaf->gain;
    if (!leaf->cats.empty()) {
      unsigned words[8] = {0};
      for (int b : leaf->cats) words[b >> 5] |= (1u << (b & 31));
      cat_off_[nid] = (int)(cat_words_.size() / 8);
      for (int w = 0; w < 8; ++w) cat_words_.push_back((int)words[w]);
    }
    const int lid = new_node();
    const int rid = new_node();
    left_[nid] = lid;
    right_[nid] = rid;
    count_[lid] = (float)CL;
    count_[rid] = (float)CR;
    value_[lid] = (float)leaf_output(GL, HL, l1, l2, max_delta);
    value_[rid] = (float)leaf_output(GR, HR, l1, l2, max_delta);

    auto lc = std::make_shared<LeafCand>();
    auto rc = std::make_shared<LeafCand>();
    lc->node_id = lid; rc->node_id = rid;
    lc->depth = rc->depth = leaf->depth + 1;
    lc->buf = rc->buf = leaf->buf ^ 1;
    lc->lo = leaf->lo; lc->hi = leaf->lo + nl;
    rc->lo = leaf->lo + nl; rc->hi = leaf->lo + m_parent;
    lc->hist = job.hist_l; rc->hist = job.hist_r;
    lc->G = GL; lc->H = HL; lc->C = CL;
    rc->G = GR; rc->H = HR; rc->C = CR;
    resolve_best(ctx, job, 0, *lc);
    resolve_best(ctx, job, 1, *rc);
    lc->seq = seq++;
    rc->seq = seq++;
    leaf->job.reset();
    leaf->hist = torch::Tensor();  // free parent histogram
    heap.push(lc);
    heap.push(rc);
    finals.push_back(lc);
    finals.push_back(rc);
    n_leaves += 1;
  }

  // leaf ordinals in node order; collect original-row-id segments from each
  // leaf's own arena buffer
  std::sort(finals.begin(), finals.end(),
            [](const CandPtr& a, const CandPtr& b) {
              return a->node_id < b->node_id;
            });
  std::vector<torch::Tensor> segs;
  std::vector<int64_t> seg_nodes;
  for (size_t i = 0; i < finals.size(); ++i) {
    leaf_idx_[finals[i]->node_id] = (int)i;
    segs.push_back(ctx.rowid[finals[i]->buf].slice(0, finals[i]->lo,
                                                   finals[i]->hi));
    seg_nodes.push_back(finals[i]->node_id);
  }
  auto leaf_rows = segs.empty()
                       ? torch::empty({0}, rows_root.options())
                       : torch::cat(segs);
  std::vector<int64_t> offs(1, 0);
  for (auto& s : segs) offs.push_back(offs.back() + s.numel());

  py::gil_scoped_acquire gil;
  auto i32 = torch::TensorOptions().dtype(torch::kInt32);
  auto f32 = torch::TensorOptions().dtype(torch::kFloat32);
  py::dict d;
  d["feature"] = torch::tensor(feature_, i32);
  d["thr_bin"] = torch::tensor(thr_bin_, i32);
  d["left"] = torch::tensor(left_, i32);
  d["right"] = torch::tensor(right_, i32);
  d["value"] = torch::tensor(value_, f32);
  d["count"] = torch::tensor(count_, f32);
  d["gain"] = torch::tensor(gain_, f32);
  d["leaf_index"] = torch::tensor(leaf_idx_, i32);
  d["leaf_rows"] = leaf_rows;
  d["leaf_offsets"] = torch::tensor(offs, torch::kInt64);
  d["leaf_nodes"] = torch::tensor(seg_nodes, torch::kInt64);
  d["cat_offset"] = torch::tensor(cat_off_, i32);
  d["cat_words"] = cat_words_.empty()
                       ? torch::zeros({0}, i32)
                       : torch::tensor(cat_words_, i32);
  return d;
}

// Standalone C++ c10d all_reduce — lets the gloo world_size=2 CPU tests
// prove the exact binding the grower uses (pg cast + allreduce + wait)
// without a GPU, and gives other native paths a GIL-free reduce.
void allreduce_native(py::object process_group, torch::Tensor t) {
  auto pg = process_group.cast<c10::intrusive_ptr<c10d::ProcessGroup>>();
  py::gil_scoped_release nogil;
  std::vector<at::Tensor> v{t};
  pg->allreduce(v)->wait();
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("grow_tree_native", &grow_tree_native,
        "native leaf-wise GBDT tree growth (numeric features, arena mode)");
  m.def("allreduce_native", &allreduce_native,
        "GIL-free c10d all_reduce through the same C++ path as the grower");
}
