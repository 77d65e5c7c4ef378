"""Validation metrics for GBDT training (distributed-aware).

Sums are all_reduced so every rank sees the same metric; AUC/NDCG are
computed on the local shard and averaged (documented approximation for
multi-rank validation sets).
"""
from __future__ import annotations

import numpy as np
import torch


def binary_logloss(p, y, w):
    p = p.clamp(1e-15, 1 - 1e-15)
    ll = -(y * torch.log(p) + (1 - y) * torch.log(1 - p))
    if w is not None:
        return float((ll * w).sum()), float(w.sum())
    return float(ll.sum()), float(ll.numel())


def auc(p: np.ndarray, y: np.ndarray) -> float:
    order = np.argsort(p, kind="stable")
    y = y[order]
    n_pos = y.sum()
    n_neg = len(y) - n_pos
    if n_pos == 0 or n_neg == 0:
        return 0.5
    ranks = np.arange(1, len(y) + 1, dtype=np.float64)
    # average ranks for ties
    ps = p[order]
    uniq, inv, cnt = np.unique(ps, return_inverse=True, return_counts=True)
    cum = np.cumsum(cnt)
    avg_rank = (cum - (cnt - 1) / 2.0)[inv]
    sum_rank_pos = avg_rank[y > 0.5].sum()
    return float((sum_rank_pos - n_pos * (n_pos + 1) / 2) / (n_pos * n_neg))


# Per-metric improvement direction (LightGBM's per-metric handling): the
# direction must come from the metric actually compared, NOT from the
# objective (an AUC-eval'd classifier is higher-better even though the
# binary objective's loss is lower-better).
_METRIC_HIGHER_BETTER = {
    "auc": True, "ndcg": True, "map": True, "accuracy": True,
    "average_precision": True, "auprc": True,
    "binary_logloss": False, "multi_logloss": False, "cross_entropy": False,
    "l1": False, "l2": False, "mae": False, "mse": False, "rmse": False,
    "mape": False, "huber": False, "fair": False, "poisson": False,
    "quantile": False, "tweedie": False, "gamma": False,
}


def metric_higher_is_better(name: str, default: bool = False) -> bool:
    """Direction of a metric by name ('ndcg@5' → 'ndcg'); falls back to
    `default` (the objective's direction) for unknown custom metrics."""
    base = str(name).split("@")[0].strip().lower()
    return _METRIC_HIGHER_BETTER.get(base, default)


def default_metrics_fn(metric_name=None):
    def fn(booster, Xv, yv, wv, comm):
        out = {}
        raw = booster.predict_raw(Xv)
        y = yv.float()
        if booster.objective == "binary":
            p = torch.sigmoid(booster.sigmoid * raw.squeeze(-1))
            s, n = binary_logloss(p, y, wv)
            t = torch.tensor([s, n], device=Xv.device)
            comm.all_reduce(t)
            out["binary_logloss"] = float(t[0] / t[1])
            # distributed note: this is the MEAN of per-rank AUCs over each
            # rank's validation shard — deterministic and identical on every
            # rank (what early stopping needs), but not the pooled global
            # AUC; pass a replicated validation set for an exact global AUC
            a = auc(p.cpu().numpy(), y.cpu().numpy())
            t = torch.tensor([a, 1.0], device=Xv.device)
            comm.all_reduce(t)
            out["auc"] = float(t[0] / t[1])
        elif booster.objective in ("multiclass", "softmax"):
            logp = torch.log_softmax(raw, dim=-1)
            nll = -logp[torch.arange(len(y), device=y.device), y.long()]
            t = torch.tensor([float(nll.sum()), float(len(y))], device=Xv.device)
            comm.all_reduce(t)
            out["multi_logloss"] = float(t[0] / t[1])
        else:
            d = raw.squeeze(-1) - y
            if wv is not None:
                t = torch.tensor([float((d * d * wv).sum()), float(wv.sum())],
                                 device=Xv.device)
            else:
                t = torch.tensor([float((d * d).sum()), float(len(y))],
                                 device=Xv.device)
            comm.all_reduce(t)
            out["l2"] = float(t[0] / t[1])
            out["rmse"] = float(np.sqrt(out["l2"]))
        if metric_name and metric_name in out:
            return {metric_name: out[metric_name], **out}
        return out
    return fn
