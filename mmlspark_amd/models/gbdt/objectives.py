"""Objective functions: per-row gradient/hessian + raw-score transforms.

Covers the reference's objective surface (params/TrainParams.scala objective
strings; ObjectiveParams / FObjTrait custom objectives): binary, multiclass,
regression (l2/l1/huber/fair/poisson/quantile/mape/tweedie), lambdarank.
Custom objective = any callable (preds, label, weight) -> (grad, hess),
the analog of FObjTrait (lightgbm/.../params/FObjTrait.scala).
"""
from __future__ import annotations

from typing import Callable, Optional, Tuple

import torch

Tensor = torch.Tensor


class Objective:
    name = "base"
    n_outputs = 1
    higher_better_metric = False

    def init_score(self, label: Tensor, weight: Optional[Tensor]) -> Tensor:
        return torch.zeros(self.n_outputs, dtype=torch.float32, device=label.device)

    def grad_hess(self, preds: Tensor, label: Tensor,
                  weight: Optional[Tensor]) -> Tuple[Tensor, Tensor]:
        raise NotImplementedError

    def transform(self, raw: Tensor) -> Tensor:
        """Raw scores -> output scale (probability etc.)."""
        return raw

    def _apply_weight(self, g, h, weight):
        if weight is not None:
            w = weight.unsqueeze(-1) if g.dim() > weight.dim() else weight
            g = g * w
            h = h * w
        return g, h


class BinaryObjective(Objective):
    """Sigmoid cross-entropy; labels in {0,1}."""
    name = "binary"

    def __init__(self, sigmoid: float = 1.0):
        self.sigmoid = sigmoid

    def init_score(self, label, weight):
        if weight is None:
            p = label.float().mean().clamp(1e-6, 1 - 1e-6)
        else:
            p = ((label.float() * weight).sum() / weight.sum()).clamp(1e-6, 1 - 1e-6)
        return torch.log(p / (1 - p)).reshape(1) / self.sigmoid

    def grad_hess(self, preds, label, weight):
        z = torch.sigmoid(self.sigmoid * preds.squeeze(-1))
        g = (z - label.float()) * self.sigmoid
        h = (z * (1 - z)).clamp_min(1e-16) * self.sigmoid * self.sigmoid
        return self._apply_weight(g.unsqueeze(-1), h.unsqueeze(-1), weight)

    def transform(self, raw):
        p1 = torch.sigmoid(self.sigmoid * raw)
        return torch.cat([1 - p1, p1], dim=-1)


class MulticlassObjective(Objective):
    """Softmax cross-entropy; one tree per class per iteration."""
    name = "multiclass"

    def __init__(self, num_class: int):
        self.n_outputs = num_class

    def init_score(self, label, weight):
        return torch.zeros(self.n_outputs, dtype=torch.float32, device=label.device)

    def grad_hess(self, preds, label, weight):
        p = torch.softmax(preds, dim=-1)
        y = torch.nn.functional.one_hot(label.long(), self.n_outputs).float()
        g = p - y
        h = (2.0 * p * (1 - p)).clamp_min(1e-16)
        return self._apply_weight(g, h, weight)

    def transform(self, raw):
        return torch.softmax(raw, dim=-1)


class RegressionL2(Objective):
    name = "regression"

    def init_score(self, label, weight):
        if weight is None:
            m = label.float().mean()
        else:
            m = (label.float() * weight).sum() / weight.sum()
        return m.reshape(1)

    def grad_hess(self, preds, label, weight):
        g = preds.squeeze(-1) - label.float()
        h = torch.ones_like(g)
        return self._apply_weight(g.unsqueeze(-1), h.unsqueeze(-1), weight)


class RegressionL1(Objective):
    name = "regression_l1"

    def init_score(self, label, weight):
        return label.float().median().reshape(1)

    def grad_hess(self, preds, label, weight):
        g = torch.sign(preds.squeeze(-1) - label.float())
        h = torch.ones_like(g)
        return self._apply_weight(g.unsqueeze(-1), h.unsqueeze(-1), weight)


class HuberObjective(Objective):
    name = "huber"

    def __init__(self, alpha: float = 0.9):
        self.alpha = alpha

    def grad_hess(self, preds, label, weight):
        d = preds.squeeze(-1) - label.float()
        g = torch.where(d.abs() <= self.alpha, d, self.alpha * torch.sign(d))
        h = torch.ones_like(g)
        return self._apply_weight(g.unsqueeze(-1), h.unsqueeze(-1), weight)


class FairObjective(Objective):
    name = "fair"

    def __init__(self, c: float = 1.0):
        self.c = c

    def grad_hess(self, preds, label, weight):
        d = preds.squeeze(-1) - label.float()
        g = self.c * d / (d.abs() + self.c)
        h = (self.c * self.c / (d.abs() + self.c) ** 2).clamp_min(1e-16)
        return self._apply_weight(g.unsqueeze(-1), h.unsqueeze(-1), weight)


class PoissonObjective(Objective):
    name = "poisson"

    def init_score(self, label, weight):
        return torch.log(label.float().mean().clamp_min(1e-8)).reshape(1)

    def grad_hess(self, preds, label, weight):
        mu = torch.exp(preds.squeeze(-1))
        g = mu - label.float()
        h = mu.clamp_min(1e-16)
        return self._apply_weight(g.unsqueeze(-1), h.unsqueeze(-1), weight)

    def transform(self, raw):
        return torch.exp(raw)


class TweedieObjective(Objective):
    name = "tweedie"

    def __init__(self, rho: float = 1.5):
        self.rho = rho

    def init_score(self, label, weight):
        return torch.log(label.float().mean().clamp_min(1e-8)).reshape(1)

    def grad_hess(self, preds, label, weight):
        x = preds.squeeze(-1)
        y = label.float()
        g = -y * torch.exp((1 - self.rho) * x) + torch.exp((2 - self.rho) * x)
        h = (-y * (1 - self.rho) * torch.exp((1 - self.rho) * x)
             + (2 - self.rho) * torch.exp((2 - self.rho) * x)).clamp_min(1e-16)
        return self._apply_weight(g.unsqueeze(-1), h.unsqueeze(-1), weight)

    def transform(self, raw):
        return torch.exp(raw)


class QuantileObjective(Objective):
    name = "quantile"

    def __init__(self, alpha: float = 0.5):
        self.alpha = alpha

    def grad_hess(self, preds, label, weight):
        d = preds.squeeze(-1) - label.float()
        g = torch.where(d >= 0, torch.full_like(d, 1 - self.alpha),
                        torch.full_like(d, -self.alpha))
        h = torch.ones_like(g)
        return self._apply_weight(g.unsqueeze(-1), h.unsqueeze(-1), weight)


class MapeObjective(Objective):
    name = "mape"

    def grad_hess(self, preds, label, weight):
        y = label.float()
        scale = 1.0 / y.abs().clamp_min(1.0)
        d = preds.squeeze(-1) - y
        g = torch.sign(d) * scale
        h = scale
        return self._apply_weight(g.unsqueeze(-1), h.unsqueeze(-1), weight)


class LambdarankObjective(Objective):
    """LambdaRank NDCG gradients over query groups.

    The ranker path of the reference (LightGBMRanker.scala:26; group column
    semantics params/LightGBMParams.scala:76). group_sizes set per-fit.
    """
    name = "lambdarank"
    higher_better_metric = True

    def __init__(self, sigmoid: float = 1.0, label_gain=None, truncation: int = 30):
        self.sigmoid = sigmoid
        self.label_gain = label_gain
        self.truncation = truncation
        self.group_sizes: Optional[Tensor] = None

    def _gains(self, label: Tensor) -> Tensor:
        if self.label_gain is not None:
            lg = torch.as_tensor(self.label_gain, dtype=torch.float32,
                                 device=label.device)
            return lg[label.long()]
        return 2.0 ** label.float() - 1.0

    def grad_hess(self, preds, label, weight):
        assert self.group_sizes is not None, "ranker requires group sizes"
        s = preds.squeeze(-1)
        g = torch.zeros_like(s)
        h = torch.zeros_like(s)
        start = 0
        for sz in self.group_sizes.tolist():
            sz = int(sz)
            if sz <= 1:
                start += sz
                continue
            sl = slice(start, start + sz)
            gi, hi = self._group_grads(s[sl], label[sl])
            g[sl] = gi
            h[sl] = hi
            start += sz
        g, h = g.unsqueeze(-1), h.unsqueeze(-1).clamp_min(1e-16)
        return self._apply_weight(g, h, weight)

    def _group_grads(self, s: Tensor, y: Tensor):
        m = s.numel()
        gains = self._gains(y)
        order = torch.argsort(s, descending=True)
        rank = torch.empty_like(order)
        rank[order] = torch.arange(m, device=s.device)
        disc = 1.0 / torch.log2(rank.float() + 2.0)
        ideal_order = torch.argsort(gains, descending=True)
        ideal_disc = 1.0 / torch.log2(torch.arange(m, device=s.device).float() + 2.0)
        idcg = (gains[ideal_order] * ideal_disc).sum().clamp_min(1e-12)
        # pairwise (i beats j where y_i > y_j)
        dy = y.unsqueeze(1) - y.unsqueeze(0)
        valid = dy > 0
        if not bool(valid.any()):
            return torch.zeros_like(s), torch.zeros_like(s)
        ds = s.unsqueeze(1) - s.unsqueeze(0)
        rho = torch.sigmoid(-self.sigmoid * ds)  # prob of mis-order
        delta_ndcg = ((gains.unsqueeze(1) - gains.unsqueeze(0)).abs()
                      * (disc.unsqueeze(1) - disc.unsqueeze(0)).abs()) / idcg
        lam = torch.where(valid, -self.sigmoid * rho * delta_ndcg,
                          torch.zeros_like(rho))
        hes = torch.where(valid, self.sigmoid * self.sigmoid * rho * (1 - rho)
                          * delta_ndcg, torch.zeros_like(rho))
        g = lam.sum(dim=1) - lam.sum(dim=0)
        h = hes.sum(dim=1) + hes.sum(dim=0)
        return g, h


class CustomObjective(Objective):
    """User fobj: callable (preds, label, weight) -> (grad, hess).

    Analog of FObjTrait (lightgbm/.../params/FObjTrait.scala) /
    updateOneIterationCustom (booster/LightGBMBooster.scala:368).
    """
    name = "custom"

    def __init__(self, fobj: Callable, base: Optional[Objective] = None):
        self.fobj = fobj
        # objective param still names the output transform (ObjectiveParams
        # semantics: fobj drives training, objective drives predict scale)
        self.base = base
        if base is not None:
            self.name = base.name
            self.n_outputs = base.n_outputs
            self.higher_better_metric = base.higher_better_metric

    def init_score(self, label, weight):
        if self.base is not None:
            return self.base.init_score(label, weight)
        return super().init_score(label, weight)

    def transform(self, raw):
        return self.base.transform(raw) if self.base is not None else raw

    def grad_hess(self, preds, label, weight):
        g, h = self.fobj(preds, label, weight)
        if g.dim() == 1:
            g, h = g.unsqueeze(-1), h.unsqueeze(-1)
        return g.float(), h.float().clamp_min(1e-16)


def make_objective(name: str, *, num_class: int = 2, sigmoid: float = 1.0,
                   alpha: float = 0.9, fair_c: float = 1.0,
                   tweedie_variance_power: float = 1.5,
                   label_gain=None, fobj: Optional[Callable] = None) -> Objective:
    if fobj is not None:
        base = None
        if name:
            base = make_objective(name, num_class=num_class, sigmoid=sigmoid,
                                  alpha=alpha, fair_c=fair_c,
                                  tweedie_variance_power=tweedie_variance_power,
                                  label_gain=label_gain)
        return CustomObjective(fobj, base)
    name = (name or "regression").lower()
    table = {
        "binary": lambda: BinaryObjective(sigmoid),
        "multiclass": lambda: MulticlassObjective(num_class),
        "softmax": lambda: MulticlassObjective(num_class),
        "regression": RegressionL2,
        "regression_l2": RegressionL2,
        "mean_squared_error": RegressionL2,
        "mse": RegressionL2,
        "l2": RegressionL2,
        "regression_l1": RegressionL1,
        "l1": RegressionL1,
        "mae": RegressionL1,
        "huber": lambda: HuberObjective(alpha),
        "fair": lambda: FairObjective(fair_c),
        "poisson": PoissonObjective,
        "tweedie": lambda: TweedieObjective(tweedie_variance_power),
        "quantile": lambda: QuantileObjective(alpha),
        "mape": MapeObjective,
        "lambdarank": lambda: LambdarankObjective(sigmoid, label_gain),
    }
    if name not in table:
        raise ValueError(f"unknown objective {name!r}")
    return table[name]()
