"""Distributed leaf-wise GBDT trainer — the MI355X rebuild of LightGBM's core.

What the reference's opaque native lib does per iteration behind
``LGBM_BoosterUpdateOneIter`` (SURVEY §2.1 "inside the native lib":
per-rank feature-histogram build, Reduce-Scatter of histograms over a TCP
ring, best-split scan, AllGather of splits, synchronized leaf-wise growth)
is re-derived here MI355X-first:

  * each rank owns a row shard in HBM as a feature-interleaved binned
    uint8 matrix (ngroups, n_rows, 4);
  * per-leaf histograms are built by the CDNA4 LDS-staged histogram kernel
    (ops/hip/gbdt_kernels.hip) and synchronized with ONE RCCL all_reduce
    over xGMI (nf × 256 × 3 floats ≈ 300 KB — latency-bound, single launch;
    the sibling histogram comes free by parent−child subtraction, so only
    the globally-smaller child is ever reduced);
  * every rank runs the identical split scan on the identical reduced
    histogram, so growth is synchronized with no split AllGather at all.

Supports gbdt / rf / dart / goss boosting, bagging, feature_fraction,
L1/L2 regularization, min_data/min_hessian/min_gain constraints, max_depth,
early stopping, validation metrics — the parameter surface of
params/LightGBMParams.scala.
"""
from __future__ import annotations

import heapq
import math
import os
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional

import numpy as np
import torch

from ...ops import backend
from ...parallel.comm import Comm
from .binning import BinMapper
from .booster import Booster
from .objectives import Objective
from .tree import Tree

NEG_INF = float("-inf")


@dataclass
class TrainConfig:
    num_iterations: int = 100
    learning_rate: float = 0.1
    num_leaves: int = 31
    max_depth: int = -1
    max_bin: int = 255
    lambda_l1: float = 0.0
    lambda_l2: float = 0.0
    min_data_in_leaf: int = 20
    min_sum_hessian_in_leaf: float = 1e-3
    min_gain_to_split: float = 0.0
    feature_fraction: float = 1.0
    bagging_fraction: float = 1.0
    bagging_freq: int = 0
    bagging_seed: int = 3        # baggingSeed (LightGBM default 3)
    boosting: str = "gbdt"  # gbdt | rf | dart | goss
    top_rate: float = 0.2       # goss
    other_rate: float = 0.1     # goss
    drop_rate: float = 0.1      # dart
    skip_drop: float = 0.5      # dart
    max_drop: int = 50          # dart
    max_delta_step: float = 0.0
    seed: int = 0
    early_stopping_round: int = 0
    first_metric_only: bool = True
    verbosity: int = -1
    is_provide_training_metric: bool = False
    metric: str = ""
    categorical_features: Optional[List[int]] = None
    parallelism: str = "data_parallel"
    top_k: int = 20
    boost_from_average: bool = True
    improvement_tolerance: float = 0.0
    pos_bagging_fraction: float = 1.0   # stratified bagging (binary labels)
    neg_bagging_fraction: float = 1.0
    bin_sample_count: int = 200_000     # binSampleCount
    max_bin_by_feature: Optional[List[int]] = None
    uniform_drop: bool = True           # dart: uniform tree-drop selection
    xgboost_dart_mode: bool = False


@dataclass
class TrainingStats:
    """Per-phase wall time — the analog of VW TrainingStats / the perf-stats
    DataFrame idea (VowpalWabbitBase.scala:27-46,464-490)."""
    hist_s: float = 0.0
    comm_s: float = 0.0
    split_s: float = 0.0
    partition_s: float = 0.0
    grad_s: float = 0.0
    eval_s: float = 0.0
    total_s: float = 0.0
    iterations: int = 0
    evals: List[Dict] = field(default_factory=list)

    def as_dict(self):
        return {k: getattr(self, k) for k in
                ("hist_s", "comm_s", "split_s", "partition_s", "grad_s",
                 "eval_s", "total_s", "iterations")}


class _Leaf:
    __slots__ = ("node_id", "rows", "hist", "G", "H", "C", "depth",
                 "gain", "feat", "bin", "GL", "HL", "CL", "cats")

    def __init__(self, node_id, rows, hist, G, H, C, depth):
        self.node_id = node_id
        self.rows = rows
        self.hist = hist
        self.G, self.H, self.C = G, H, C
        self.depth = depth
        self.gain = NEG_INF
        self.cats = None

    def __lt__(self, other):  # max-heap via negated gain at push site
        return False


def _leaf_output(G, H, cfg: TrainConfig) -> float:
    g = abs(G) - cfg.lambda_l1
    if g <= 0:
        return 0.0
    denom = H + cfg.lambda_l2
    if denom <= 0:
        # a leaf whose (quantized) hessians all rounded to zero with no L2 —
        # LightGBM outputs 0 below min_sum_hessian rather than diverging
        return 0.0
    w = -math.copysign(g, G) / denom
    if cfg.max_delta_step > 0:
        w = max(-cfg.max_delta_step, min(cfg.max_delta_step, w))
    return w


class TreeGrower:
    """Grows one tree leaf-wise on the local shard with global histogram sync."""

    def __init__(self, binned_i4: torch.Tensor, n_features: int,
                 cfg: TrainConfig, comm: Comm, stats: TrainingStats,
                 bin_mapper: BinMapper, n_global: Optional[int] = None):
        self.binned = binned_i4
        self.nf = n_features
        self.nf_pad = binned_i4.shape[0] * 4
        self.cfg = cfg
        self.comm = comm
        self.stats = stats
        self.bin_mapper = bin_mapper
        self.device = binned_i4.device
        # fixed-point integer histograms on GPU (9x faster LDS atomics;
        # bit-exact all_reduce + sibling subtraction)
        self.fixed = self.device.type == "cuda"
        if self.fixed:
            # pair adjacent uchar4 planes into one u64 plane: the hist kernel
            # then fetches ONE cacheline per gathered row per feature-chunk
            # block (plane-major pairs are ~40 MB apart → two fetches before)
            g32 = binned_i4.view(torch.int32).reshape(binned_i4.shape[0], -1)
            G = g32.shape[0]
            if G % 2:
                g32 = torch.cat([g32, torch.zeros_like(g32[:1])])
            lo = g32[0::2].to(torch.int64) & 0xFFFFFFFF
            hi = g32[1::2].to(torch.int64) & 0xFFFFFFFF
            self.binned_pair = (lo | (hi << 32)).contiguous()
            self.nf_pad = self.binned_pair.shape[0] * 8
            self.tail_bytes = G * 4 - (self.binned_pair.shape[0] - 1) * 8
        self.n_global = n_global or binned_i4.shape[1]
        self.scale_g = 1.0
        self.scale_h = 1.0
        self.cat_features = list(getattr(bin_mapper, "categorical", []) or [])
        self.cat_smooth = 10.0

    _sync_timers = bool(os.environ.get("MMLSPARK_AMD_SYNC_TIMERS"))

    def set_scales(self, grad: torch.Tensor, hess: torch.Tensor):
        """Per-tree fixed-point scales from GLOBAL max|g|, max h (identical on
        every rank: one tiny all_reduce)."""
        if not self.fixed:
            return
        mx = torch.stack([grad.abs().max(), hess.max()])
        self.comm.all_reduce(mx, op="max")
        gmax = max(float(mx[0]), 1e-12)
        hmax = max(float(mx[1]), 1e-12)
        self.scale_g = (2.0 ** 61) / (max(self.n_global, 1) * gmax)
        self.scale_h = (2.0 ** 24) / hmax  # chunk ≤ 2^19 rows → fits 44 bits

    @property
    def voting(self) -> bool:
        """Voting-parallel: reduce only globally-voted top-K
        feature histograms instead of all of them."""
        return (getattr(self.cfg, "parallelism", "data_parallel")
                == "voting_parallel" and self.comm.is_distributed)

    def _hist(self, rows, grad, hess, reduce=True):
        t0 = time.perf_counter()
        if self.fixed:
            h = backend.hist_build_fixed_pair(self.binned_pair, rows, grad,
                                              hess, self.cfg.max_bin,
                                              self.tail_bytes,
                                              self.scale_g, self.scale_h)
        else:
            h = backend.hist_build(self.binned, rows, grad, hess,
                                   self.cfg.max_bin)
        if self._sync_timers and self.device.type == "cuda":
            torch.cuda.synchronize()
        t1 = time.perf_counter()
        self.stats.hist_s += t1 - t0
        if reduce:
            if self.voting:
                self._voting_reduce(h)
            else:
                self.comm.all_reduce(h)
            self.stats.comm_s += time.perf_counter() - t1
        return h

    def _voting_reduce(self, h: torch.Tensor):
        """Local top-K vote → all_gather of candidate feature ids →
        all_reduce of the union's histograms only (LightGBM voting_parallel,
        LightGBMConstants TopK=20).  Non-union features get their counts
        zeroed so the split scan can never pick a locally-reduced feature."""
        k = min(getattr(self.cfg, "top_k", 20), self.nf)
        histf = self._to_float_hist(h.unsqueeze(0))[0]
        # per-feature gains for the vote
        g = histf[:, :, 0].cumsum(1)
        hh = histf[:, :, 1].cumsum(1)
        gains = (g[:, :-1] ** 2 / (hh[:, :-1] + 1e-6)
                 + (g[:, -1:] - g[:, :-1]) ** 2
                 / (hh[:, -1:] - hh[:, :-1] + 1e-6)).amax(dim=1)
        gains[self.nf:] = float("-inf")
        local_top = torch.topk(gains, k).indices.to(torch.int64)
        cands = torch.cat(self.comm.all_gather(local_top.contiguous()))
        union = torch.unique(cands)
        sub = h.index_select(0, union).contiguous()
        self.comm.all_reduce(sub)
        keep = torch.zeros(h.shape[0], dtype=torch.bool, device=h.device)
        keep[union] = True
        h[~keep] = 0  # counts 0 ⇒ min_data_in_leaf rejects these features
        h[union] = sub

    def _to_float_hist(self, hists: torch.Tensor) -> torch.Tensor:
        """Stacked histograms → float32 real units for the split scan."""
        if not self.fixed:
            return hists
        # float64 from the start: a float32 intermediate here would round
        # 1/scale differently than the fixed-scan kernel's double inv_g and
        # make native- vs Python-grown trees differ in final ulps
        inv = torch.tensor([1.0 / self.scale_g, 1.0 / self.scale_h, 1.0],
                           dtype=torch.float64, device=hists.device)
        return (hists.double() * inv).float()

    def _sums(self, hist: torch.Tensor):
        """(G, H, C) totals of one histogram (feature 0 owns every row)."""
        s = hist[0].sum(dim=0)
        if self.fixed:
            return (float(s[0]) / self.scale_g, float(s[1]) / self.scale_h,
                    float(s[2]))
        return float(s[0]), float(s[1]), float(s[2])

    def _scan(self, hists, feat_mask):
        """Fused split scan on stacked histograms; identical on all ranks
        (input is the reduced histogram). One kernel pair + one readback.
        Categorical features use a torch-side sorted one-vs-rest scan
        (LightGBM categorical semantics) and compete with the numeric best."""
        t0 = time.perf_counter()
        cfg = self.cfg
        histsf = self._to_float_hist(hists)
        num_mask = feat_mask
        if self.cat_features:
            num_mask = (feat_mask.clone() if feat_mask is not None else
                        torch.ones(self.nf_pad, dtype=torch.bool,
                                   device=hists.device))
            num_mask[self.cat_features] = False
        out = backend.split_scan(
            histsf, cfg.max_bin, cfg.lambda_l1,
            cfg.lambda_l2, float(cfg.min_data_in_leaf),
            cfg.min_sum_hessian_in_leaf, cfg.min_gain_to_split, self.nf,
            num_mask).cpu()
        res = []
        for hi, row in enumerate(out.tolist()):
            bg, bf, bb, gl, hl, cl = row
            best = [bg, int(bf), int(bb), gl, hl, cl, None]
            if self.cat_features:
                cat = self._cat_scan(histsf[hi], feat_mask)
                if cat is not None and cat[0] > best[0]:
                    best = cat
            res.append(tuple(best))
        self.stats.split_s += time.perf_counter() - t0
        return res

    def _cat_scan(self, histf, feat_mask):
        """Sorted one-vs-rest categorical split over present categories.

        Deterministic: float32 numpy arithmetic with a lexsort tie-break by
        bin id — the C++ arena grower (gbdt_grower.cpp cat_scan_host)
        replicates this exactly so native and Python growers produce
        bit-identical trees."""
        cfg = self.cfg
        best = None
        hcpu = histf.cpu()
        for f in self.cat_features:
            if feat_mask is not None and not bool(feat_mask[f]):
                continue
            h = hcpu[f].numpy().astype(np.float32)  # (nb, 3): g, h, count
            present = np.nonzero(h[:, 2] > 0)[0]
            if len(present) < 2:
                continue
            g = h[present, 0]
            hh = h[present, 1]
            c = h[present, 2]
            ratio = (g / (hh + np.float32(self.cat_smooth))).astype(np.float32)
            order = np.lexsort((present, ratio))  # stable: ratio, then bin id
            GL = np.cumsum(g[order], dtype=np.float32)
            HL = np.cumsum(hh[order], dtype=np.float32)
            CL = np.cumsum(c[order], dtype=np.float32)
            G, H, C = GL[-1], HL[-1], CL[-1]

            def sc(Gs, Hs):
                Ga = np.maximum(np.abs(Gs) - np.float32(cfg.lambda_l1),
                                np.float32(0))
                return Ga * Ga / (Hs + np.float32(cfg.lambda_l2)
                                  + np.float32(1e-32))

            gain = sc(GL, HL) + sc(G - GL, H - HL) - sc(
                np.float32(G), np.float32(H))
            valid = ((CL >= cfg.min_data_in_leaf)
                     & (C - CL >= cfg.min_data_in_leaf)
                     & (HL >= cfg.min_sum_hessian_in_leaf)
                     & (H - HL >= cfg.min_sum_hessian_in_leaf))
            gain = np.where(valid, gain, np.float32(NEG_INF))
            gain[-1] = NEG_INF
            k = int(gain.argmax())
            bg = float(gain[k])
            if not np.isfinite(bg):
                continue
            if best is None or bg > best[0]:
                cats = present[order[: k + 1]].tolist()
                best = [bg, int(f), 0, float(GL[k]), float(HL[k]),
                        float(CL[k]), cats]
        return best

    def _best_split(self, hist, feat_mask):
        return self._scan(hist.unsqueeze(0), feat_mask)[0]

    def _partition(self, leaf) -> tuple:
        """Split a leaf's row list by its committed split (stable order)."""
        t0 = time.perf_counter()
        if leaf.cats is not None:
            gq, j = leaf.feat // 4, leaf.feat % 4
            bins = self.binned[gq, leaf.rows.long(), j]
            cats_t = torch.tensor(leaf.cats, dtype=bins.dtype,
                                  device=bins.device)
            mask = torch.isin(bins, cats_t)
            rows_l, rows_r = leaf.rows[mask], leaf.rows[~mask]
        else:
            # counts above 2^24 lose exactness through the f32 scan
            known = (int(leaf.CL) if (self.fixed
                                      and not self.comm.is_distributed
                                      and leaf.C < 1.6e7)
                     else -1)
            rows_l, rows_r = backend.partition_rows(
                self.binned, leaf.rows, leaf.feat, leaf.bin,
                known_left=known)
        self.stats.partition_s += time.perf_counter() - t0
        return rows_l, rows_r

    def _grow_native(self, rows_root, grad, hess, feat_mask):
        """Whole leaf-wise loop in the C++ driver (ops/hip/gbdt_grower.cpp)
        — one Python call per tree instead of ~6 per split."""
        from ...ops import _hip_grower
        cfg = self.cfg
        # the per-split histogram all_reduce runs INSIDE the C++ grower via
        # the c10d ProcessGroup C++ API — no GIL hop on the critical path
        pg = self.comm.native_group()
        # categorical features: numeric scan masks them out; the grower's
        # host-side cat scan handles them (filtered by this iteration's
        # feature mask, mirroring _scan/_cat_scan)
        cats_active = [f for f in self.cat_features
                       if feat_mask is None or bool(feat_mask[f])]
        num_mask = feat_mask
        if self.cat_features:
            num_mask = (feat_mask.clone() if feat_mask is not None else
                        torch.ones(self.nf_pad, dtype=torch.bool,
                                   device=self.device))
            num_mask[self.cat_features] = False
        cats_t = (torch.tensor(cats_active, dtype=torch.int64)
                  if cats_active else None)
        t0 = time.perf_counter()
        d = _hip_grower.grow_tree_native(
            self.binned, self.binned_pair, rows_root.contiguous(), grad, hess,
            cfg.max_bin,
            self.nf, self.scale_g, self.scale_h, cfg.lambda_l1, cfg.lambda_l2,
            float(cfg.min_data_in_leaf), cfg.min_sum_hessian_in_leaf,
            cfg.min_gain_to_split, cfg.max_delta_step, cfg.num_leaves,
            cfg.max_depth, num_mask, cats_t, self.cat_smooth, pg,
            self.comm.is_distributed or bool(os.environ.get(
                "MMLSPARK_AMD_FORCE_DIST_GROWER")))
        self.stats.hist_s += time.perf_counter() - t0
        feature = d["feature"].numpy()
        thr_bin = d["thr_bin"].numpy()
        cat_offset = d["cat_offset"].numpy()
        threshold = np.array(
            [float("nan") if cat_offset[i] >= 0 else
             self.bin_mapper.bin_upper_value(int(f), int(b)) if f >= 0 else 0.0
             for i, (f, b) in enumerate(zip(feature, thr_bin))],
            dtype=np.float32)
        tree = Tree(feature, threshold, thr_bin, d["left"].numpy(),
                    d["right"].numpy(), d["value"].numpy(),
                    d["count"].numpy(), d["gain"].numpy(),
                    d["leaf_index"].numpy(), shrinkage=1.0,
                    cat_offset=cat_offset,
                    cat_words=d["cat_words"].numpy().view(np.uint32))
        offs = d["leaf_offsets"].tolist()
        nodes = d["leaf_nodes"].tolist()
        leaves = []
        for i, nid in enumerate(nodes):
            lf = _Leaf(int(nid), d["leaf_rows"][offs[i]:offs[i + 1]], None,
                       0, 0, 0, 0)
            leaves.append(lf)
        return tree, leaves

    native_ok = True  # sparse grower overrides (no arena path yet)

    def grow(self, rows_root: torch.Tensor, grad: torch.Tensor,
             hess: torch.Tensor, feat_mask) -> (Tree, List):
        if (self.native_ok and self.fixed and not self.voting
                and not os.environ.get(
                    "MMLSPARK_AMD_NO_NATIVE_GROWER")):
            try:
                from ...ops import _hip_grower  # noqa: F401
                self.set_scales(grad, hess)
                return self._grow_native(rows_root, grad, hess, feat_mask)
            except ImportError:
                pass
        cfg = self.cfg
        # node arrays (grown dynamically)
        feature, threshold, thr_bin = [], [], []
        left, right, value, count, gain_arr, leaf_idx = [], [], [], [], [], []
        cat_off, cat_words = [], []

        def new_node():
            feature.append(-1)
            threshold.append(0.0)
            thr_bin.append(0)
            left.append(-1)
            right.append(-1)
            value.append(0.0)
            count.append(0.0)
            gain_arr.append(0.0)
            leaf_idx.append(-1)
            cat_off.append(-1)
            return len(feature) - 1

        self.set_scales(grad, hess)
        root_hist = self._hist(rows_root, grad, hess)
        if self.voting:
            t = torch.stack([grad[rows_root.long()].sum(),
                             hess[rows_root.long()].sum(),
                             torch.tensor(float(rows_root.numel()),
                                          device=self.device)])
            self.comm.all_reduce(t)
            G, H, C = float(t[0]), float(t[1]), float(t[2])
        else:
            G, H, C = self._sums(root_hist)
        root = _Leaf(new_node(), rows_root, root_hist, G, H, C, 0)
        (root.gain, root.feat, root.bin, root.GL, root.HL, root.CL,
         root.cats) = self._best_split(root_hist, feat_mask)
        count[root.node_id] = C
        value[root.node_id] = _leaf_output(G, H, cfg)

        heap = [(-root.gain, 0, root)]
        seq = 1
        n_leaves = 1
        final_leaves = [root]

        while n_leaves < cfg.num_leaves and heap:
            ngain, _, leaf = heapq.heappop(heap)
            if -ngain <= cfg.min_gain_to_split or not np.isfinite(-ngain):
                continue
            if cfg.max_depth > 0 and leaf.depth >= cfg.max_depth:
                continue
            final_leaves.remove(leaf)

            rows_l, rows_r = self._partition(leaf)

            GL, HL, CL = leaf.GL, leaf.HL, leaf.CL
            GR, HR, CR = leaf.G - GL, leaf.H - HL, leaf.C - CL
            if self.voting:
                # voting mode reduces a per-leaf feature union, so the parent
                # histogram is not globally complete — build both children
                hist_l = self._hist(rows_l, grad, hess)
                hist_r = self._hist(rows_r, grad, hess)
            else:
                # reduce only the globally-smaller child; sibling by subtraction
                left_small = CL <= CR
                small_rows = rows_l if left_small else rows_r
                hist_small = self._hist(small_rows, grad, hess)
                hist_big = leaf.hist - hist_small
                hist_l, hist_r = ((hist_small, hist_big) if left_small
                                  else (hist_big, hist_small))

            nid = leaf.node_id
            feature[nid] = leaf.feat
            thr_bin[nid] = leaf.bin
            if leaf.cats is not None:
                words = [0] * 8
                for b in leaf.cats:
                    words[b >> 5] |= 1 << (b & 31)
                cat_off[nid] = len(cat_words) // 8
                cat_words.extend(words)
                threshold[nid] = float("nan")  # categorical marker for dumps
            else:
                threshold[nid] = self.bin_mapper.bin_upper_value(leaf.feat,
                                                                 leaf.bin)
            gain_arr[nid] = -ngain
            lid, rid = new_node(), new_node()
            left[nid], right[nid] = lid, rid
            count[lid], count[rid] = CL, CR
            value[lid] = _leaf_output(GL, HL, cfg)
            value[rid] = _leaf_output(GR, HR, cfg)

            lc = _Leaf(lid, rows_l, hist_l, GL, HL, CL, leaf.depth + 1)
            rc = _Leaf(rid, rows_r, hist_r, GR, HR, CR, leaf.depth + 1)
            leaf.hist = None  # free parent histogram
            # both children in ONE fused scan + readback
            pair = self._scan(torch.stack([hist_l, hist_r]), feat_mask)
            for ch, res in zip((lc, rc), pair):
                (ch.gain, ch.feat, ch.bin, ch.GL, ch.HL, ch.CL, ch.cats) = res
                heapq.heappush(heap, (-ch.gain, seq, ch))
                seq += 1
                final_leaves.append(ch)
            n_leaves += 1

        # leaf ordinals in node-creation order
        for i, lf in enumerate(sorted(final_leaves, key=lambda l: l.node_id)):
            leaf_idx[lf.node_id] = i
            lf.hist = None

        tree = Tree(feature, threshold, thr_bin, left, right, value, count,
                    gain_arr, leaf_idx, shrinkage=1.0,
                    cat_offset=cat_off, cat_words=cat_words)
        return tree, final_leaves


class SparseTreeGrower(TreeGrower):
    """Leaf-wise growth over a binned CSR shard (models/gbdt/sparse.py).

    The histogram covers stored entries only; each feature's implicit zeros
    are recovered by exact integer subtraction from leaf totals, so the
    multi-rank all_reduce stays bit-exact.  Replaces the dense-only binned
    matrix for high-dimensional sparse workloads (the reference's
    LGBM_DatasetCreateFromCSR path, DatasetAggregator.scala:442)."""

    native_ok = False  # python-loop grower; arena mode is dense-only

    def __init__(self, shard, n_features: int, cfg: TrainConfig, comm: Comm,
                 stats: TrainingStats, bin_mapper: BinMapper,
                 n_global: Optional[int] = None):
        self.shard = shard
        self.binned = None
        self.nf = n_features
        self.nf_pad = n_features
        self.cfg = cfg
        self.comm = comm
        self.stats = stats
        self.bin_mapper = bin_mapper
        self.device = shard.device
        self.fixed = True  # int64 fixed point on CPU and GPU alike
        self.n_global = n_global or shard.shape[0]
        self.scale_g = 1.0
        self.scale_h = 1.0
        self.cat_features = []
        self.cat_smooth = 10.0
        self._gq = None
        self._hq = None

    def set_scales(self, grad: torch.Tensor, hess: torch.Tensor):
        mx = torch.stack([grad.abs().max(), hess.max()])
        self.comm.all_reduce(mx, op="max")
        gmax = max(float(mx[0]), 1e-12)
        hmax = max(float(mx[1]), 1e-12)
        self.scale_g = (2.0 ** 61) / (max(self.n_global, 1) * gmax)
        self.scale_h = (2.0 ** 24) / hmax
        # quantize once per tree — every rank rounds identically
        self._gq = torch.round(grad.double() * self.scale_g).to(torch.int64)
        self._hq = torch.round(hess.double() * self.scale_h).to(torch.int64)

    def _hist(self, rows, grad, hess, reduce=True):
        t0 = time.perf_counter()
        sh = self.shard
        h, tot = backend.csr_hist_fixed_tot(sh.indptr, sh.col, sh.binv,
                                            self._gq, self._hq, rows,
                                            self.nf, self.cfg.max_bin)
        # implicit zeros: exact integer leaf totals − per-feature stored sums
        tot = tot.to(h.device)
        corr = tot.unsqueeze(0) - h.sum(dim=1)  # (nf, 3)
        ar = torch.arange(self.nf, device=h.device)
        h[ar, sh.zero_bin.long()] += corr
        if self._sync_timers and self.device.type == "cuda":
            torch.cuda.synchronize()
        t1 = time.perf_counter()
        self.stats.hist_s += t1 - t0
        if reduce:
            if self.voting:
                self._voting_reduce(h)
            else:
                self.comm.all_reduce(h)
            self.stats.comm_s += time.perf_counter() - t1
        return h

    def _partition(self, leaf) -> tuple:
        t0 = time.perf_counter()
        sh = self.shard
        # counts above 2^24 lose exactness through the f32 scan
        known = (int(leaf.CL) if (not self.comm.is_distributed
                                  and leaf.C < 1.6e7)
                 else -1)
        rows_l, rows_r = backend.csr_partition_rows(
            sh.indptr, sh.col, sh.binv, leaf.rows, leaf.feat,
            int(sh.zero_bin[leaf.feat]), leaf.bin, known_left=known)
        self.stats.partition_s += time.perf_counter() - t0
        return rows_l, rows_r


def predict_tree_binned(tree: Tree, binned_i4: torch.Tensor,
                        device) -> torch.Tensor:
    """Leaf value per local row by binned traversal (used by DART drops)."""
    n = binned_i4.shape[1]
    feat = torch.from_numpy(tree.feature).to(device).long()
    thrb = torch.from_numpy(tree.thr_bin).to(device)
    lft = torch.from_numpy(tree.left).to(device).long()
    rgt = torch.from_numpy(tree.right).to(device).long()
    val = torch.from_numpy(tree.value).to(device)
    catoff = torch.from_numpy(tree.cat_offset).to(device).long()
    catw = (torch.from_numpy(tree.cat_words.view(np.int32).copy()).to(device)
            if len(tree.cat_words) else None)
    idx = torch.zeros(n, dtype=torch.long, device=device)
    active = feat[idx] >= 0
    flat = binned_i4.permute(0, 2, 1).reshape(-1, n)  # (nf_pad, n) view by feature
    while bool(active.any()):
        f = feat[idx].clamp(min=0)
        bins = flat[f, torch.arange(n, device=device)]
        go_left = bins <= thrb[idx]
        if catw is not None:
            off = catoff[idx]
            is_cat = off >= 0
            if bool(is_cat.any()):
                b = bins.long()
                widx = (off.clamp(min=0) * 8 + (b >> 5)).clamp(
                    max=catw.numel() - 1)
                w32 = catw[widx].long() & 0xFFFFFFFF
                bits = ((w32 >> (b & 31)) & 1).bool()
                go_left = torch.where(is_cat, bits, go_left)
        nxt = torch.where(go_left, lft[idx], rgt[idx])
        idx = torch.where(active, nxt, idx)
        active = feat[idx] >= 0
    return val[idx]


def _goss_sample(grad, hess, cfg: TrainConfig, gen) -> torch.Tensor:
    """GOSS: keep top_rate by |g|, sample other_rate of the rest, amplify."""
    n = grad.shape[0]
    a, b = cfg.top_rate, cfg.other_rate
    g_abs = grad.abs().sum(dim=-1) if grad.dim() > 1 else grad.abs()
    n_top = max(1, int(a * n))
    n_rest = max(1, int(b * n))
    order = torch.argsort(g_abs, descending=True)
    top = order[:n_top]
    rest_pool = order[n_top:]
    if rest_pool.numel() > 0:
        perm = torch.randperm(rest_pool.numel(), generator=gen,
                              device="cpu").to(grad.device)
        rest = rest_pool[perm[:n_rest]]
        amp = (1.0 - a) / max(b, 1e-12)
        grad[rest] *= amp
        hess[rest] *= amp
        rows = torch.cat([top, rest])
    else:
        rows = top
    return rows.to(torch.int32).sort().values


class TrainingSession:
    """Step-level training state: one ``step()`` = one boosting iteration
    (the bench.py "step"; also the iteration-level checkpoint boundary)."""

    def __init__(self, X: torch.Tensor, y: torch.Tensor, cfg: TrainConfig,
                 objective: Objective, comm: Comm,
                 weight: Optional[torch.Tensor] = None,
                 group_sizes: Optional[torch.Tensor] = None,
                 feature_names: Optional[List[str]] = None,
                 init_booster: Optional[Booster] = None,
                 binned_cache=None, init_score: Optional[torch.Tensor] = None):
        self.X, self.y, self.cfg, self.objective = X, y, cfg, objective
        self.comm, self.weight = comm, weight
        self.feature_names = feature_names
        self.stats = TrainingStats()
        self.device = X.device
        self.n, self.nf = X.shape
        self.K = objective.n_outputs
        if hasattr(objective, "group_sizes"):
            objective.group_sizes = group_sizes

        from .sparse import CsrMatrix, bin_csr, fit_bin_mapper_csr
        self.is_sparse = isinstance(X, CsrMatrix)
        if self.is_sparse:
            if cfg.categorical_features:
                raise ValueError("categorical features need dense input "
                                 "(set matrixType='dense')")
            if cfg.boosting == "dart":
                raise ValueError("dart boosting needs dense input "
                                 "(set matrixType='dense')")
        if binned_cache is not None:
            self.bin_mapper, self.binned = binned_cache
        elif self.is_sparse:
            self.bin_mapper = fit_bin_mapper_csr(
                X, n_bins=cfg.max_bin, comm=comm, seed=cfg.seed,
                sample_size=cfg.bin_sample_count)
            self.binned = bin_csr(X, self.bin_mapper)
        else:
            self.bin_mapper = BinMapper.fit(X, n_bins=cfg.max_bin, comm=comm,
                                            seed=cfg.seed,
                                            sample_size=cfg.bin_sample_count,
                                            categorical=cfg.categorical_features,
                                            max_bin_by_feature=cfg.max_bin_by_feature)
            self.binned = self.bin_mapper.transform(X)

        device, n, K = self.device, self.n, self.K
        if init_booster is not None and init_booster.trees:
            base = torch.from_numpy(init_booster.base_score).to(device)
            self.preds = init_booster.predict_raw(X)
            self.trees: List[Tree] = list(init_booster.trees)
            self.tree_w: List[float] = list(init_booster.tree_weights)
        else:
            base = self._global_init_score()
            if base.numel() < K:
                base = base.expand(K).contiguous()
            self.preds = base.unsqueeze(0).expand(n, K).clone()
            if init_score is not None:  # initScoreCol warm-start margins
                sc = init_score.to(device).float()
                self.preds += sc.unsqueeze(-1) if sc.dim() == 1 else sc
            self.trees = []
            self.tree_w = []
        self.base = base
        n_global = n
        if comm.is_distributed:
            t = torch.tensor([float(n)], device=device)
            comm.all_reduce(t)
            n_global = int(t[0])
        grower_cls = SparseTreeGrower if self.is_sparse else TreeGrower
        self.grower = grower_cls(self.binned, self.nf, cfg, comm, self.stats,
                                 self.bin_mapper, n_global=n_global)
        self.gen = torch.Generator(device="cpu")
        self.all_rows = torch.arange(n, dtype=torch.int32, device=device)
        self.n_start_trees = len(self.trees)
        self.it = 0

    def _global_init_score(self):
        y, weight, device = self.y, self.weight, self.device
        obj, comm = self.objective, self.comm
        if not self.cfg.boost_from_average:  # boostFromAverage=False
            return torch.zeros(1, device=device)
        if comm.is_distributed:
            # the stored base_score must be IDENTICAL on every rank — a
            # shard-local init would make per-rank models differ even though
            # the reduced-histogram trees agree
            w = weight if weight is not None else torch.ones_like(
                y, dtype=torch.float32)
            sums = torch.stack([(y.float() * w).sum(), w.sum()]).to(device)
            comm.all_reduce(sums)
            mean = sums[0] / sums[1].clamp_min(1e-12)
            if obj.name == "binary":
                p = mean.clamp(1e-6, 1 - 1e-6)
                sig = getattr(obj, "sigmoid", 1.0)
                return (torch.log(p / (1 - p)) / sig).reshape(1)
            if obj.name in ("regression", "regression_l2"):
                return mean.reshape(1)
            if obj.name in ("poisson", "tweedie"):
                return torch.log(mean.clamp_min(1e-8)).reshape(1)
            # every other objective (l1 median, quantile, custom fobj):
            # average the shard-local inits — approximate but identical on
            # all ranks, and the first boosting round absorbs the error
            s0 = obj.init_score(y, weight).to(device).float()
            comm.all_reduce(s0)
            return s0 / comm.world_size
        return obj.init_score(y, weight).to(device)

    # ------------------------------------------------------------------ step
    def step(self):
        cfg, gen, K, device = self.cfg, self.gen, self.K, self.device
        trees, tree_w, preds = self.trees, self.tree_w, self.preds
        it = self.it
        rf_mode = cfg.boosting == "rf"
        dart_mode = cfg.boosting == "dart"
        goss_mode = cfg.boosting == "goss"

        # ---- gradients ----------------------------------------------------
        t0 = time.perf_counter()
        dropped = []
        if dart_mode and trees[self.n_start_trees:]:
            gen.manual_seed(cfg.seed * 7919 + it)
            if float(torch.rand(1, generator=gen)) >= cfg.skip_drop:
                cand = list(range(self.n_start_trees, len(trees)))
                if cfg.uniform_drop:
                    pdrop = torch.full((len(cand),), cfg.drop_rate)
                else:  # drop probability proportional to tree weight
                    tw = torch.tensor([abs(tree_w[t]) for t in cand])
                    pdrop = (cfg.drop_rate * len(cand)
                             * tw / tw.sum().clamp_min(1e-12)).clamp(0, 1)
                mask = torch.rand(len(cand), generator=gen) < pdrop
                dropped = [cand[i] for i in range(len(cand)) if bool(mask[i])]
                dropped = dropped[: cfg.max_drop]
        if dropped:
            drop_contrib = torch.zeros_like(preds)
            for t in dropped:
                k = (t - self.n_start_trees) % K
                drop_contrib[:, k] += tree_w[t] * predict_tree_binned(
                    trees[t], self.binned, device)
            preds_used = preds - drop_contrib
        else:
            preds_used = preds
        if rf_mode:
            preds_used = self.base.unsqueeze(0).expand(self.n, K)
        grad, hess = self.objective.grad_hess(preds_used, self.y, self.weight)
        self.stats.grad_s += time.perf_counter() - t0

        # ---- row sampling (baggingSeed drives bagging/GOSS draws) -----------
        gen.manual_seed(cfg.bagging_seed * 104729 + it * 31 + self.comm.rank)
        if goss_mode and it >= 1:
            rows_root = _goss_sample(grad, hess, cfg, gen)
        elif (cfg.pos_bagging_fraction < 1.0 or cfg.neg_bagging_fraction < 1.0) \
                and cfg.bagging_freq > 0 and it % cfg.bagging_freq == 0:
            # posBaggingFraction/negBaggingFraction: stratified row bagging
            pos = (self.y > 0).cpu()
            keep = torch.rand(self.n, generator=gen) < torch.where(
                pos, torch.tensor(cfg.pos_bagging_fraction),
                torch.tensor(cfg.neg_bagging_fraction))
            rows_root = keep.nonzero(as_tuple=True)[0].to(
                device, torch.int32).sort().values
            if rows_root.numel() == 0:
                rows_root = self.all_rows
        elif (cfg.bagging_freq > 0 and cfg.bagging_fraction < 1.0
              and it % cfg.bagging_freq == 0) or rf_mode:
            frac = cfg.bagging_fraction if cfg.bagging_fraction < 1.0 else 0.632
            m = max(1, int(frac * self.n))
            perm = torch.randperm(self.n, generator=gen)[:m].to(device)
            rows_root = perm.to(torch.int32).sort().values
        else:
            rows_root = self.all_rows

        # ---- feature sampling (same seed on every rank) ---------------------
        feat_mask = None
        if cfg.feature_fraction < 1.0:
            gen_f = torch.Generator(device="cpu")
            gen_f.manual_seed(cfg.seed * 524287 + it)
            kf = max(1, int(cfg.feature_fraction * self.nf))
            sel = torch.randperm(self.nf, generator=gen_f)[:kf]
            feat_mask = torch.zeros(self.grower.nf_pad, dtype=torch.bool,
                                    device=device)
            feat_mask[sel.to(device)] = True

        # ---- one tree per output class --------------------------------------
        new_trees = []
        for k in range(K):
            tree, leaves = self.grower.grow(rows_root, grad[:, k].contiguous(),
                                            hess[:, k].contiguous(), feat_mask)
            tree.shrinkage = 1.0 if rf_mode else cfg.learning_rate
            if not rf_mode:
                # one fused update: concat leaf row lists + repeat leaf values
                t0 = time.perf_counter()
                live = [lf for lf in leaves if lf.rows.numel()]
                if live:
                    all_rows = torch.cat([lf.rows for lf in live]).long()
                    sizes = [lf.rows.numel() for lf in live]
                    vals = torch.tensor(
                        [float(tree.value[lf.node_id]) * tree.shrinkage
                         for lf in live], device=device)
                    expanded = torch.repeat_interleave(
                        vals, torch.tensor(sizes, device=device))
                    preds[:, k].index_add_(0, all_rows, expanded)
                self.stats.partition_s += time.perf_counter() - t0
            new_trees.append(tree)

        if dropped:
            # DART normalization — default: dropped ×k/(k+1), new ×1/(k+1);
            # xgboostDartMode: dropped ×k/(k+lr), new ×lr/(k+lr)
            kdrop = len(dropped)
            if cfg.xgboost_dart_mode:
                factor = kdrop / (kdrop + cfg.learning_rate)
                new_scale = cfg.learning_rate / (kdrop + cfg.learning_rate)
            else:
                factor = kdrop / (kdrop + 1.0)
                new_scale = 1.0 / (kdrop + 1.0)
            for t in dropped:
                k = (t - self.n_start_trees) % K
                delta = (factor - 1.0) * tree_w[t]
                preds[:, k] += delta * predict_tree_binned(trees[t], self.binned,
                                                           device)
                tree_w[t] *= factor
            for k, tr in enumerate(new_trees):
                excess = (1.0 - new_scale) * tr.shrinkage
                preds[:, k] -= excess * predict_tree_binned(tr, self.binned, device)
                tr.shrinkage *= new_scale

        for tr in new_trees:
            trees.append(tr)
            tree_w.append(1.0)
        self.stats.iterations += 1
        self.it += 1

    # --------------------------------------------------------------- snapshot
    def booster(self) -> Booster:
        tw = list(self.tree_w)
        if self.cfg.boosting == "rf" and self.stats.iterations:
            tw = [1.0 / self.stats.iterations] * len(tw)
        return Booster(list(self.trees), self.objective.name, self.K,
                       self.base.cpu().numpy(), self.nf, self.feature_names,
                       self.bin_mapper,
                       sigmoid=getattr(self.objective, "sigmoid", 1.0),
                       tree_weights=np.array(tw, dtype=np.float32))


def load_checkpoint(checkpoint_dir: str):
    """(booster, completed_iterations) from an iteration-level checkpoint,
    or None.  The elastic-restart resume point (the reference's recovery
    story is Spark barrier gang restart + modelString warm start,
    LightGBMBase.scala:46-61; here a restarted gang resumes mid-training)."""
    path = os.path.join(checkpoint_dir, "checkpoint.json")
    if not os.path.exists(path):
        return None
    import json as _json
    with open(path) as f:
        d = _json.load(f)
    return Booster.load_from_string(d["model"]), int(d["iteration"])


def _save_checkpoint(checkpoint_dir: str, booster: Booster, iteration: int):
    import json as _json
    os.makedirs(checkpoint_dir, exist_ok=True)
    tmp = os.path.join(checkpoint_dir, ".checkpoint.tmp")
    with open(tmp, "w") as f:
        _json.dump({"iteration": iteration,
                    "model": booster.save_to_string()}, f)
    os.replace(tmp, os.path.join(checkpoint_dir, "checkpoint.json"))


def train_booster(X: torch.Tensor, y: torch.Tensor, cfg: TrainConfig,
                  objective: Objective, comm: Comm,
                  weight: Optional[torch.Tensor] = None,
                  group_sizes: Optional[torch.Tensor] = None,
                  feature_names: Optional[List[str]] = None,
                  valid_sets: Optional[List[tuple]] = None,
                  init_booster: Optional[Booster] = None,
                  metrics_fn=None,
                  binned_cache=None, init_score=None,
                  checkpoint_dir: Optional[str] = None,
                  checkpoint_every: int = 0) -> (Booster, TrainingStats):
    """Full training loop with eval + early stopping over TrainingSession.

    With checkpoint_dir + checkpoint_every, rank 0 writes an atomic
    iteration-level checkpoint every k iterations and a fresh call resumes
    from it — the fault-tolerance piece a restarted gang needs."""
    t_start = time.perf_counter()
    start_it = 0
    if checkpoint_dir:
        # resume state must be IDENTICAL on every rank or collective counts
        # diverge and the gang deadlocks — rank 0's checkpoint view wins
        # (checkpoint_dir may be node-local; rank>0 may see nothing)
        if comm.is_distributed:
            ck0 = load_checkpoint(checkpoint_dir) if comm.rank == 0 else None
            obj = comm.all_gather_object(
                None if ck0 is None
                else (ck0[0].save_to_string(), ck0[1]))[0]
            if obj is not None:
                init_booster = Booster.load_from_string(obj[0])
                start_it = obj[1]
        else:
            ck = load_checkpoint(checkpoint_dir)
            if ck is not None:
                init_booster, start_it = ck
    session = TrainingSession(X, y, cfg, objective, comm, weight=weight,
                              group_sizes=group_sizes,
                              feature_names=feature_names,
                              init_booster=init_booster,
                              binned_cache=binned_cache,
                              init_score=init_score)
    stats = session.stats
    best_score = None
    best_iter = -1
    rounds_no_improve = 0

    for it in range(start_it, cfg.num_iterations):
        session.step()
        if (checkpoint_dir and checkpoint_every > 0
                and (it + 1) % checkpoint_every == 0 and comm.rank == 0):
            _save_checkpoint(checkpoint_dir, session.booster(), it + 1)
        if metrics_fn is not None and (valid_sets or cfg.is_provide_training_metric):
            t0 = time.perf_counter()
            booster_now = session.booster()
            entry = {"iteration": it}
            if cfg.is_provide_training_metric:
                # per-iteration TRAIN metrics (isProvideTrainingMetric,
                # TrainUtils.scala:117-128 logs train eval each iteration)
                entry["training"] = metrics_fn(booster_now, X, y, weight,
                                               comm)
            score = None
            score_name = None
            for vi, (Xv, yv, wv) in enumerate(valid_sets or []):
                m = metrics_fn(booster_now, Xv, yv, wv, comm)
                entry[f"valid_{vi}"] = m
                if score is None and m:
                    score_name, score = next(iter(m.items()))
            stats.evals.append(entry)
            stats.eval_s += time.perf_counter() - t0
            if score is not None and cfg.early_stopping_round > 0:
                # direction keyed off the metric actually compared (per-metric
                # table like LightGBM), not the objective's loss direction
                from .metrics import metric_higher_is_better
                higher_better = metric_higher_is_better(
                    score_name, objective.higher_better_metric)
                tol = cfg.improvement_tolerance  # improvementTolerance
                better = (best_score is None
                          or (score > best_score + tol
                              if higher_better
                              else score < best_score - tol))
                if better:
                    best_score, best_iter = score, it
                    rounds_no_improve = 0
                else:
                    rounds_no_improve += 1
                    if rounds_no_improve >= cfg.early_stopping_round:
                        break

    stats.total_s = time.perf_counter() - t_start
    booster = session.booster()
    booster.best_iteration = best_iter
    return booster, stats
