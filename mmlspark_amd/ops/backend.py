"""Device-op dispatch: HIP/CDNA4 extension on GPU tensors, torch reference on CPU.

The HIP extension is built IN-TREE (mmlspark_amd/ops/_hip_ops*.so) by
``python setup.py build_ext --inplace`` / ``__graft_entry__.build()`` so the
.so travels to the GPU box with the repo snapshot.  On a CUDA/ROCm tensor the
extension is REQUIRED — a missing .so raises instead of silently falling back
to eager torch (the round-end check records which native libraries the GPU
processes actually loaded).
"""
from __future__ import annotations

import torch

from . import cpu_ref

_EXT = None
_EXT_ERR: str = ""


def _load_ext():
    global _EXT, _EXT_ERR
    if _EXT is not None:
        return _EXT
    try:
        from . import _hip_ops  # built in-tree
        _EXT = _hip_ops
    except ImportError as e:
        _EXT_ERR = str(e)
        _EXT = False
    return _EXT


def hip_available() -> bool:
    return bool(_load_ext()) and torch.cuda.is_available()


def _require_ext():
    ext = _load_ext()
    if not ext:
        raise RuntimeError(
            "mmlspark_amd HIP extension is required for GPU tensors but is not "
            "built (import error: %s). Run `python setup.py build_ext --inplace` "
            "(PYTORCH_ROCM_ARCH=gfx950)." % _EXT_ERR)
    return ext


# --------------------------------------------------------------------------- GBDT
def hist_build(binned_i4, rows, grad, hess, n_bins):
    if binned_i4.is_cuda:
        return _require_ext().hist_build(binned_i4, rows, grad, hess, n_bins)
    return cpu_ref.hist_build(binned_i4, rows, grad, hess, n_bins)


def hist_build_fixed_pair(binned_pair, rows, grad, hess, n_bins, tail_bytes,
                          scale_g, scale_h):
    """Fixed-point histogram over PAIRED planes ((npairs, n) int64: two
    uchar4 feature groups per element — one cacheline per gathered row).
    tail_bytes = valid feature-bytes in the last pair (zero-pad skipped).
    GPU-only (the pairing exists only on the device path)."""
    return _require_ext().hist_build_fixed_pair(binned_pair, rows, grad,
                                                hess, n_bins, tail_bytes,
                                                scale_g, scale_h)


def hist_build_fixed(binned_i4, rows, grad, hess, n_bins, scale_g, scale_h):
    """Fixed-point int64 histogram (GPU; 9x faster integer LDS atomics).
    Returns (nf_pad, n_bins, 3) int64: [g*scale_g, h*scale_h, count]."""
    if binned_i4.is_cuda:
        return _require_ext().hist_build_fixed(binned_i4, rows, grad, hess,
                                               n_bins, scale_g, scale_h)
    h = cpu_ref.hist_build(binned_i4, rows, grad, hess, n_bins)
    scale = torch.tensor([scale_g, scale_h, 1.0])
    return (h.double() * scale).round().long()


def partition_rows(binned_i4, rows, feature, threshold_bin, known_left=-1):
    if binned_i4.is_cuda:
        # ordered 3-kernel partition; ONE device→host sync, or zero when the
        # caller already knows the left count (single-rank training)
        return _require_ext().partition_rows(binned_i4, rows.contiguous(),
                                             feature, threshold_bin,
                                             known_left)
    return cpu_ref.partition_rows(binned_i4, rows, feature, threshold_bin)


def predict_forest(node_feature, node_threshold, node_left, node_right,
                   node_value, tree_offsets, X, n_outputs,
                   tree_weights=None, start_tree=0, num_iteration=-1,
                   cat_offset=None, cat_words=None):
    if X.is_cuda:
        n_trees = tree_offsets.numel() - 1
        end_tree = n_trees if num_iteration < 0 else min(
            n_trees, start_tree + num_iteration * n_outputs)
        if tree_weights is None:
            tree_weights = torch.ones(n_trees, dtype=torch.float32, device=X.device)
        return _require_ext().predict_forest(
            node_feature, node_threshold, node_left, node_right, node_value,
            tree_offsets, tree_weights, X, n_outputs, start_tree, end_tree,
            cat_offset, cat_words)
    return cpu_ref.predict_forest(node_feature, node_threshold, node_left,
                                  node_right, node_value, tree_offsets, X,
                                  n_outputs, tree_weights, start_tree,
                                  num_iteration, cat_offset, cat_words)


def predict_leaf(node_feature, node_threshold, node_left, node_right,
                 node_leaf_index, tree_offsets, X,
                 cat_offset=None, cat_words=None):
    if X.is_cuda:
        return _require_ext().predict_leaf(
            node_feature, node_threshold, node_left, node_right,
            node_leaf_index, tree_offsets, X, cat_offset, cat_words)
    return cpu_ref.predict_leaf(node_feature, node_threshold, node_left,
                                node_right, node_leaf_index, tree_offsets, X,
                                cat_offset, cat_words)


def split_scan(hists, n_bins, l1, l2, min_data, min_hess, min_gain, nf_real,
               feat_mask=None):
    if hists.is_cuda:
        return _require_ext().split_scan(hists.contiguous(), n_bins, l1, l2,
                                         min_data, min_hess, min_gain,
                                         nf_real, feat_mask)
    return cpu_ref.split_scan(hists, n_bins, l1, l2, min_data, min_hess,
                              min_gain, nf_real, feat_mask)


def bin_matrix(X, upper_bounds, n_bins):
    if X.is_cuda:
        return _require_ext().bin_matrix(X, upper_bounds, n_bins)
    return cpu_ref.bin_matrix(X, upper_bounds, n_bins)


# --------------------------------------------------------------------------- VW
def vw_sgd_minibatch(indices, values, offsets, labels, weights_tbl, adaptive_tbl,
                     lr, l2, power_t, loss: str, ex_weight=None,
                     normalize_tbl=None, invariant=False):
    if weights_tbl.is_cuda:
        return _require_ext().vw_sgd_minibatch(
            indices, values, offsets, labels, weights_tbl, adaptive_tbl,
            lr, l2, power_t, {"squared": 0, "logistic": 1, "hinge": 2}[loss],
            ex_weight, normalize_tbl, invariant)
    from ..models.vw import sgd_ref
    return sgd_ref.vw_sgd_minibatch(indices, values, offsets, labels,
                                    weights_tbl, adaptive_tbl, lr, l2,
                                    power_t, loss, ex_weight, normalize_tbl,
                                    invariant)


def vw_predict(indices, values, offsets, weights_tbl):
    if weights_tbl.is_cuda:
        return _require_ext().vw_predict(indices, values, offsets, weights_tbl)
    from ..models.vw import sgd_ref
    return sgd_ref.vw_predict(indices, values, offsets, weights_tbl)


# --------------------------------------------------------------- sparse CSR
def csr_hist_fixed(indptr, col, binv, gq, hq, rows, nf, n_bins):
    """Fixed-point histogram over STORED entries of `rows` from a binned CSR
    shard; (nf, n_bins, 3) int64.  Implicit-zero correction is the caller's
    (exact integer subtraction from leaf totals)."""
    if binv.is_cuda:
        return _require_ext().csr_hist_fixed(indptr, col, binv, gq, hq,
                                             rows.contiguous(), nf, n_bins)
    return cpu_ref.csr_hist_fixed(indptr, col, binv, gq, hq, rows, nf, n_bins)


def csr_gather_bins(indptr, col, binv, rows, feature, zero_bin):
    """Per-row bin of `feature` (missing → zero_bin) from a binned CSR shard."""
    if binv.is_cuda:
        return _require_ext().csr_gather_bins(indptr, col, binv,
                                              rows.contiguous(), feature,
                                              zero_bin).long()
    return cpu_ref.csr_gather_bins(indptr, col, binv, rows, feature, zero_bin)


def csr_hist_fixed_tot(indptr, col, binv, gq, hq, rows, nf, n_bins):
    """Histogram over stored entries PLUS exact integer leaf totals
    (sum gq, sum hq, count) in one launch — (hist, tot).  GPU: the
    wave-cooperative v2 kernel; CPU: reference + torch sums."""
    if binv.is_cuda:
        return _require_ext().csr_hist_fixed_tot(indptr, col, binv, gq, hq,
                                                 rows.contiguous(), nf,
                                                 n_bins)
    h = cpu_ref.csr_hist_fixed(indptr, col, binv, gq, hq, rows, nf, n_bins)
    r = rows.long()
    tot = torch.stack([gq[r].sum(), hq[r].sum(),
                       torch.tensor(int(rows.numel()), dtype=torch.int64)])
    return h, tot


def csr_partition_rows(indptr, col, binv, rows, feature, zero_bin,
                       threshold_bin, known_left=-1):
    """Stable ordered partition of a CSR shard's row list by one feature's
    bin (missing → zero_bin): fused predicate + 3-kernel partition on GPU,
    gather + boolean masks on CPU."""
    if binv.is_cuda:
        return _require_ext().csr_partition_rows(
            indptr, col, binv, rows.contiguous(), feature, zero_bin,
            threshold_bin, known_left)
    bins = cpu_ref.csr_gather_bins(indptr, col, binv, rows, feature, zero_bin)
    mask = bins <= threshold_bin
    return rows[mask], rows[~mask]
