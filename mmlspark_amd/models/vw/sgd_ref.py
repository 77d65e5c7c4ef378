"""Torch CPU reference for the VW sparse-SGD kernels (minibatch/hogwild
semantics matching vw_kernels.hip; numerics reference for GPU tests)."""
from __future__ import annotations

import torch

EPS = 1e-10


def _dloss(loss: str, pred: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
    if loss == "squared":
        return pred - y
    if loss == "logistic":  # y in {-1, +1}
        return -y * torch.sigmoid(-y * pred)
    if loss == "hinge":
        return torch.where(y * pred < 1.0, -y, torch.zeros_like(y))
    raise ValueError(loss)


def vw_predict(indices, values, offsets, w_tbl):
    n_ex = offsets.numel() - 1
    counts = offsets[1:] - offsets[:-1]
    seg = torch.repeat_interleave(torch.arange(n_ex, device=indices.device),
                                  counts)
    contrib = w_tbl[indices.long()] * values
    out = torch.zeros(n_ex, dtype=torch.float32, device=indices.device)
    out.index_add_(0, seg, contrib)
    return out


def vw_sgd_minibatch(indices, values, offsets, labels, w_tbl, g_tbl, lr, l2,
                     power_t, loss: str, ex_weight=None, s_tbl=None):
    """One pass over the minibatch: adaptive (AdaGrad) sparse updates.
    Collisions accumulate like the GPU kernel's atomics (index_add)."""
    n_ex = offsets.numel() - 1
    counts = offsets[1:] - offsets[:-1]
    seg = torch.repeat_interleave(torch.arange(n_ex, device=indices.device),
                                  counts)
    il = indices.long()
    preds = vw_predict(indices, values, offsets, w_tbl)
    gl = _dloss(loss, preds, labels)
    if ex_weight is not None:
        gl = gl * ex_weight
    g = gl[seg] * values + l2 * w_tbl[il]
    g_tbl.index_add_(0, il, g * g)
    G = g_tbl[il]
    if power_t == 0.5:
        scale = torch.rsqrt(G + EPS)
    else:
        scale = (G + EPS) ** (-power_t)
    if s_tbl is not None:  # --normalized: running max|x| per weight
        s_tbl.scatter_reduce_(0, il, values.abs(), reduce="amax")
        sn = s_tbl[il].clamp_min(EPS)
        scale = scale / sn
    w_tbl.index_add_(0, il, -lr * g * scale)
    return preds
