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


def _invariant_dp(loss: str, pred, y, h_eta):
    """Prediction-space move of the importance-weight-invariant update
    (Karampatziakis & Langford, VW --invariant): integrate the per-example
    gradient flow dp/dh = -eta * dloss(p) exactly over importance weight h,
    so a weight-h update equals h sequential weight-1 updates in the small-h
    limit and NEVER overshoots the label/margin for any h."""
    if loss == "squared":  # p(h) = y + (p0-y) exp(-h eta)
        return (y - pred) * (-torch.expm1(-h_eta))
    if loss == "logistic":
        # q = y*p obeys e^q + q = e^{q0} + q0 + h*eta; solve d = q-q0 >= 0
        q0 = y * pred
        A = torch.exp(q0.clamp(max=30.0))
        d = torch.where(q0 > 30.0, h_eta * torch.exp(-q0),
                        torch.log1p(h_eta / (A + 1.0)))
        for _ in range(8):  # Newton on g(d)=A expm1(d)+d-h_eta (convex)
            g = A * torch.expm1(d) + d - h_eta
            d = (d - g / (A * torch.exp(d) + 1.0)).clamp_min(0.0)
        return y * d
    if loss == "hinge":  # move to the margin, never past it
        q0 = y * pred
        d = torch.minimum(h_eta, (1.0 - q0).clamp_min(0.0))
        return y * d
    raise ValueError(loss)


def vw_sgd_minibatch(indices, values, offsets, labels, w_tbl, g_tbl, lr, l2,
                     power_t, loss: str, ex_weight=None, s_tbl=None,
                     invariant=False):
    """One pass over the minibatch: adaptive (AdaGrad) sparse updates.
    Collisions accumulate like the GPU kernel's atomics (index_add).
    invariant=True applies VW's importance-weight-aware closed-form update
    (safe for large importance weights)."""
    n_ex = offsets.numel() - 1
    counts = offsets[1:] - offsets[:-1]
    seg = torch.repeat_interleave(torch.arange(n_ex, device=indices.device),
                                  counts)
    il = indices.long()
    preds = vw_predict(indices, values, offsets, w_tbl)
    gl0 = _dloss(loss, preds, labels)
    gl = gl0 * ex_weight if ex_weight is not None else gl0
    g = gl[seg] * values + l2 * w_tbl[il]
    if s_tbl is not None:  # --normalized: running max|x| per weight
        s_tbl.scatter_reduce_(0, il, values.abs(), reduce="amax")
    if invariant:
        # per-coordinate scale with G-before-update + x^2 proxy (VW's
        # pred_per_update sensitivity), then a prediction-space solve
        Gp = g_tbl[il] + values * values
        scale = torch.rsqrt(Gp + EPS) if power_t == 0.5 \
            else (Gp + EPS) ** (-power_t)
        if s_tbl is not None:
            scale = scale / s_tbl[il].clamp_min(EPS)
        xs2 = values * values * scale
        x_norm = torch.zeros(n_ex, dtype=torch.float32,
                             device=indices.device)
        x_norm.index_add_(0, seg, xs2)  # sensitivity sum x_i^2 s_i
        h = ex_weight if ex_weight is not None else torch.ones_like(labels)
        dp = _invariant_dp(loss, preds, labels, h * lr * x_norm)
        g_tbl.index_add_(0, il, g * g)
        k = (dp / x_norm.clamp_min(EPS))[seg]
        w_tbl.index_add_(0, il, (k * values - lr * l2 * w_tbl[il]) * scale)
        return preds
    g_tbl.index_add_(0, il, g * g)
    G = g_tbl[il]
    if power_t == 0.5:
        scale = torch.rsqrt(G + EPS)
    else:
        scale = (G + EPS) ** (-power_t)
    if s_tbl is not None:
        sn = s_tbl[il].clamp_min(EPS)
        scale = scale / sn
    w_tbl.index_add_(0, il, -lr * g * scale)
    return preds
