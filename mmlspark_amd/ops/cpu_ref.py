"""Pure-torch CPU reference implementations of every device op.

These are (a) the CPU execution path for GPU-less environments and
(b) the numerics reference the HIP kernels are tested against
(tests compare HIP output to these in fp32).
"""
from __future__ import annotations

import torch


def hist_build(binned_i4: torch.Tensor, rows: torch.Tensor, grad: torch.Tensor,
               hess: torch.Tensor, n_bins: int) -> torch.Tensor:
    """Per-(feature, bin) gradient/hessian/count sums over the given rows.

    binned_i4: (ngroups, n_rows, 4) uint8 — feature-interleaved binned data,
               feature f lives at [f // 4, :, f % 4].
    rows:      (m,) int32/int64 row indices of the leaf.
    grad/hess: (n_rows,) float32.
    returns:   (ngroups*4, n_bins, 3) float32 [sum_grad, sum_hess, count].
    """
    ng = binned_i4.shape[0]
    nf = ng * 4
    r = rows.long()
    b = binned_i4[:, r, :].permute(0, 2, 1).reshape(nf, -1).long()  # (nf, m)
    g = grad[r].unsqueeze(0).expand(nf, -1)
    h = hess[r].unsqueeze(0).expand(nf, -1)
    hist = torch.zeros(nf, n_bins, 3, dtype=torch.float32, device=binned_i4.device)
    hist[:, :, 0].scatter_add_(1, b, g)
    hist[:, :, 1].scatter_add_(1, b, h)
    hist[:, :, 2].scatter_add_(1, b, torch.ones_like(g))
    return hist


def partition_rows(binned_i4: torch.Tensor, rows: torch.Tensor, feature: int,
                   threshold_bin: int):
    """Split rows into (left, right): left ⇔ bin[feature] <= threshold_bin."""
    g, j = feature // 4, feature % 4
    bins = binned_i4[g, rows.long(), j]
    mask = bins <= threshold_bin
    return rows[mask], rows[~mask]


def _go_left(xv, idx, node_threshold, cat_offset, cat_words):
    """Vectorized branch decision incl. categorical bitset nodes."""
    gl = (xv <= node_threshold[idx]) | torch.isnan(xv)
    if cat_offset is not None and cat_words is not None and cat_words.numel():
        off = cat_offset[idx].long()
        is_cat = off >= 0
        if bool(is_cat.any()):
            b = torch.nan_to_num(xv, nan=0.0).round().long().clamp(0, 255)
            widx = (off.clamp(min=0) * 8 + (b >> 5)).clamp(
                max=cat_words.numel() - 1)
            w32 = cat_words[widx].long() & 0xFFFFFFFF
            bits = (w32 >> (b & 31)) & 1
            in_range = (xv >= 0) & (xv < 256) & ~torch.isnan(xv)
            cat_left = torch.where(in_range, bits.bool(),
                                   torch.ones_like(bits, dtype=torch.bool))
            gl = torch.where(is_cat, cat_left, gl)
    return gl


def predict_forest(node_feature: torch.Tensor, node_threshold: torch.Tensor,
                   node_left: torch.Tensor, node_right: torch.Tensor,
                   node_value: torch.Tensor, tree_offsets: torch.Tensor,
                   X: torch.Tensor, n_outputs: int,
                   tree_weights: torch.Tensor = None,
                   start_tree: int = 0, num_iteration: int = -1,
                   cat_offset: torch.Tensor = None,
                   cat_words: torch.Tensor = None) -> torch.Tensor:
    """Sum of per-tree leaf values over the ensemble. Vectorized traversal.

    Flattened node arrays (concatenated trees); interior node: feature >= 0,
    go left iff X[:, feature] <= threshold or X is NaN (missing → left);
    leaf: feature == -1, value = node_value.
    Trees are laid out round-robin over outputs: tree t contributes to
    output t % n_outputs (LightGBM multiclass convention).
    returns (n, n_outputs) float32 raw scores (no base score added).
    """
    n = X.shape[0]
    n_trees = tree_offsets.numel() - 1
    end_tree = n_trees if num_iteration < 0 else min(
        n_trees, start_tree + num_iteration * n_outputs)
    T = end_tree - start_tree
    if T <= 0 or n == 0:
        return torch.zeros(n, n_outputs, dtype=torch.float32, device=X.device)
    # traverse ALL trees simultaneously: (n, T) node cursors, one vectorized
    # step per tree level instead of a python loop per tree
    bases = tree_offsets[start_tree:end_tree].long().to(X.device)  # (T,)
    idx = bases.unsqueeze(0).expand(n, T).contiguous()
    rows = torch.arange(n, device=X.device).unsqueeze(1).expand(n, T)
    active = node_feature[idx] >= 0
    while bool(active.any()):
        f = node_feature[idx].clamp(min=0).long()
        xv = X[rows, f]
        go_left = _go_left(xv, idx, node_threshold, cat_offset, cat_words)
        nxt = torch.where(go_left, node_left[idx], node_right[idx]).long()
        idx = torch.where(active, nxt + bases.unsqueeze(0), idx)
        active = node_feature[idx] >= 0
    vals = node_value[idx]
    if tree_weights is not None:
        vals = vals * tree_weights[start_tree:end_tree].unsqueeze(0)
    out = torch.zeros(n, n_outputs, dtype=torch.float32, device=X.device)
    if n_outputs == 1:
        out[:, 0] = vals.sum(dim=1)
    else:
        cls = (torch.arange(start_tree, end_tree, device=X.device)
               % n_outputs)
        out.index_add_(1, cls, vals)
    return out


def predict_leaf(node_feature, node_threshold, node_left, node_right,
                 node_leaf_index, tree_offsets, X,
                 cat_offset=None, cat_words=None) -> torch.Tensor:
    """Per-tree leaf index for each row: (n, n_trees) int32."""
    n = X.shape[0]
    n_trees = tree_offsets.numel() - 1
    if n_trees == 0 or n == 0:
        return torch.zeros(n, n_trees, dtype=torch.int32, device=X.device)
    bases = tree_offsets[:-1].long().to(X.device)
    idx = bases.unsqueeze(0).expand(n, n_trees).contiguous()
    rows = torch.arange(n, device=X.device).unsqueeze(1).expand(n, n_trees)
    active = node_feature[idx] >= 0
    while bool(active.any()):
        f = node_feature[idx].clamp(min=0).long()
        xv = X[rows, f]
        go_left = _go_left(xv, idx, node_threshold, cat_offset, cat_words)
        nxt = torch.where(go_left, node_left[idx], node_right[idx]).long()
        idx = torch.where(active, nxt + bases.unsqueeze(0), idx)
        active = node_feature[idx] >= 0
    return node_leaf_index[idx].to(torch.int32)


def bin_matrix(X: torch.Tensor, upper_bounds: torch.Tensor,
               n_bins: int) -> torch.Tensor:
    """Quantile-bin a dense (n, nf) matrix into the (ngroups, n, 4) i4 layout.

    upper_bounds: (nf, n_bins-1) float32 ascending per-feature boundaries
    (+inf padded). bin = searchsorted(bounds, x) so x <= bounds[b] → bin<=b.
    NaN maps to bin 0.
    """
    n, nf = X.shape
    ng = (nf + 3) // 4
    Xc = torch.nan_to_num(X, nan=-float("inf"))
    bins = torch.searchsorted(upper_bounds.contiguous(),
                              Xc.t().contiguous(), right=False)
    bins = bins.clamp(max=n_bins - 1).to(torch.uint8)  # (nf, n)
    out = torch.zeros(ng, n, 4, dtype=torch.uint8, device=X.device)
    for f in range(nf):
        out[f // 4, :, f % 4] = bins[f]
    return out


def split_scan(hists: torch.Tensor, n_bins: int, l1: float, l2: float,
               min_data: float, min_hess: float, min_gain: float,
               nf_real: int, feat_mask=None) -> torch.Tensor:
    """Best split per histogram. hists (n_hists, nf_pad, n_bins, 3) ->
    (n_hists, 6) {gain, feature, bin, GL, HL, CL}. Matches split_scan_k."""
    g = hists[..., 0]
    h = hists[..., 1]
    c = hists[..., 2]
    GL = g.cumsum(-1)
    HL = h.cumsum(-1)
    CL = c.cumsum(-1)
    G = GL[..., -1:]
    H = HL[..., -1:]
    C = CL[..., -1:]
    GR, HR, CR = G - GL, H - HL, C - CL

    def sc(Gs, Hs):
        Ga = (Gs.abs() - l1).clamp_min(0)
        return Ga * Ga / (Hs + l2 + 1e-32)

    gain = sc(GL, HL) + sc(GR, HR) - sc(G, H)
    valid = ((CL >= min_data) & (CR >= min_data)
             & (HL >= min_hess) & (HR >= min_hess))
    neg = torch.full_like(gain, float("-inf"))
    gain = torch.where(valid, gain, neg)
    gain[..., -1] = float("-inf")
    if feat_mask is not None:
        gain[:, ~feat_mask, :] = float("-inf")
    gain[:, nf_real:, :] = float("-inf")
    per_f, per_bin = gain.max(dim=-1)           # (nh, nf)
    bf = per_f.argmax(dim=-1)                   # (nh,)
    nh = hists.shape[0]
    ar = torch.arange(nh, device=hists.device)
    bb = per_bin[ar, bf]
    out = torch.stack([
        per_f[ar, bf], bf.float(), bb.float(),
        GL[ar, bf, bb], HL[ar, bf, bb], CL[ar, bf, bb]], dim=-1)
    return out


# ------------------------------------------------------------------ sparse CSR
def _expand_csr(indptr: torch.Tensor, rows: torch.Tensor):
    """(entry_indices, row_position_per_entry) for the given rows."""
    rows = rows.long()
    starts = indptr[rows]
    counts = indptr[rows + 1] - starts
    total = int(counts.sum())
    if total == 0:
        z = torch.zeros(0, dtype=torch.int64, device=indptr.device)
        return z, z
    seg = torch.repeat_interleave(starts, counts)
    off = torch.arange(total, device=indptr.device)
    bounds = torch.repeat_interleave(torch.cumsum(counts, 0) - counts, counts)
    rpos = torch.repeat_interleave(
        torch.arange(rows.numel(), device=indptr.device), counts)
    return seg + (off - bounds), rpos


def csr_hist_fixed(indptr, col, binv, gq, hq, rows, nf, n_bins):
    """Fixed-point histogram over stored CSR entries of `rows`
    (implicit zeros are corrected by the caller from exact leaf totals).
    Returns (nf, n_bins, 3) int64 — matches csr_hist_fixed_k."""
    e, rpos = _expand_csr(indptr, rows)
    hist = torch.zeros(nf * n_bins, 3, dtype=torch.int64, device=binv.device)
    if e.numel():
        r = rows.long()[rpos]
        flat = col[e].long() * n_bins + binv[e].long()
        src = torch.stack([gq[r], hq[r], torch.ones_like(gq[r])], dim=1)
        hist.index_add_(0, flat, src)
    return hist.view(nf, n_bins, 3)


def csr_gather_bins(indptr, col, binv, rows, feature: int, zero_bin: int):
    """Bin of `feature` for each row (missing → zero_bin); matches
    csr_gather_bin_k."""
    e, rpos = _expand_csr(indptr, rows)
    out = torch.full((rows.numel(),), int(zero_bin), dtype=torch.int64,
                     device=binv.device)
    if e.numel():
        m = col[e].long() == int(feature)
        out[rpos[m]] = binv[e[m]].long()
    return out
