"""Small dense solvers for the explainers.

Parity: LassoRegression coordinate descent (core/.../explainers/LassoRegression.scala:10)
and LeastSquaresRegression normal equations (LeastSquaresRegression.scala),
both re-expressed on torch so per-row solves can batch on device.
"""
from __future__ import annotations

from typing import NamedTuple, Optional

import torch


class RegressionResult(NamedTuple):
    coefficients: torch.Tensor
    intercept: float
    r_squared: float
    loss: float


def _standardize(X, y, sample_weight):
    w = sample_weight / sample_weight.sum()
    xm = (X * w.unsqueeze(1)).sum(0)
    ym = (y * w).sum()
    Xc = X - xm
    yc = y - ym
    xs = torch.sqrt((Xc * Xc * w.unsqueeze(1)).sum(0)).clamp_min(1e-12)
    return Xc / xs, yc, xm, ym, xs, w


def lasso_regression(X: torch.Tensor, y: torch.Tensor, alpha: float,
                     sample_weight: Optional[torch.Tensor] = None,
                     max_iter: int = 200, tol: float = 1e-6) -> RegressionResult:
    """Weighted lasso via cyclic coordinate descent on standardized features."""
    n, d = X.shape
    if sample_weight is None:
        sample_weight = torch.ones(n, dtype=X.dtype, device=X.device)
    Xs, yc, xm, ym, xs, w = _standardize(X, y, sample_weight)
    beta = torch.zeros(d, dtype=X.dtype, device=X.device)
    wX = w.unsqueeze(1) * Xs
    # precompute per-feature weighted squared norms (=1 after standardization)
    z = (wX * Xs).sum(0).clamp_min(1e-12)
    resid = yc - Xs @ beta
    for _ in range(max_iter):
        max_delta = 0.0
        for j in range(d):
            bj = float(beta[j])
            rho = float((wX[:, j] * (resid + Xs[:, j] * bj)).sum())
            bnew = torch.sign(torch.tensor(rho)) * max(abs(rho) - alpha, 0.0) / z[j]
            bnew = float(bnew)
            if bnew != bj:
                resid = resid + Xs[:, j] * (bj - bnew)
                beta[j] = bnew
                max_delta = max(max_delta, abs(bnew - bj))
        if max_delta < tol:
            break
    coef = beta / xs
    intercept = float(ym - (coef * xm).sum())
    loss = float((w * resid * resid).sum())
    var = float((w * yc * yc).sum())
    r2 = 1.0 - loss / max(var, 1e-12)
    return RegressionResult(coef, intercept, r2, loss)


def least_squares_regression(X: torch.Tensor, y: torch.Tensor,
                             sample_weight: Optional[torch.Tensor] = None,
                             l2: float = 1e-8,
                             fit_intercept: bool = True) -> RegressionResult:
    """Weighted least squares via normal equations."""
    n, d = X.shape
    if sample_weight is None:
        sample_weight = torch.ones(n, dtype=X.dtype, device=X.device)
    if fit_intercept:
        Xa = torch.cat([X, torch.ones(n, 1, dtype=X.dtype, device=X.device)], 1)
    else:
        Xa = X
    W = sample_weight
    A = Xa.t() @ (W.unsqueeze(1) * Xa)
    A += l2 * torch.eye(A.shape[0], dtype=X.dtype, device=X.device)
    b = Xa.t() @ (W * y)
    sol = torch.linalg.solve(A, b)
    coef = sol[:d]
    intercept = float(sol[d]) if fit_intercept else 0.0
    resid = y - Xa @ sol
    loss = float((W * resid * resid).sum())
    ym = (W * y).sum() / W.sum()
    var = float((W * (y - ym) ** 2).sum())
    r2 = 1.0 - loss / max(var, 1e-12)
    return RegressionResult(coef, intercept, r2, loss)


def constrained_kernel_shap_solve(Z: torch.Tensor, v: torch.Tensor,
                                  weights: torch.Tensor, v_null: float,
                                  v_full: float) -> torch.Tensor:
    """KernelSHAP WLS with the efficiency constraint sum(phi) = v_full - v_null
    (eliminating the last feature — Lundberg's standard reduction)."""
    m = Z.shape[1]
    if m == 1:
        return torch.tensor([v_full - v_null], dtype=Z.dtype, device=Z.device)
    zl = Z[:, -1]
    Zr = Z[:, :-1] - zl.unsqueeze(1)
    yr = v - v_null - zl * (v_full - v_null)
    res = least_squares_regression(Zr, yr, weights, fit_intercept=False)
    phi_rest = res.coefficients
    phi_last = (v_full - v_null) - phi_rest.sum()
    return torch.cat([phi_rest, phi_last.reshape(1)])


def batched_kernel_shap_solve(Z: torch.Tensor, V: torch.Tensor,
                              weights: torch.Tensor, v_null: torch.Tensor,
                              v_full: torch.Tensor) -> torch.Tensor:
    """Solve the constrained KernelSHAP WLS for MANY rows sharing one
    coalition design: factor A = Zr'WZr once, apply to every row/class.

    Z: (nZ, m); V: (R, nZ) model outputs per row; v_null: scalar or (R,);
    v_full: (R,).  Returns (R, m) phis."""
    m = Z.shape[1]
    R = V.shape[0]
    if m == 1:
        return (v_full - v_null).reshape(R, 1)
    zl = Z[:, -1]                      # (nZ,)
    Zr = Z[:, :-1] - zl.unsqueeze(1)   # (nZ, m-1)
    Zw = Zr * weights.unsqueeze(1)
    A = Zw.t() @ Zr                    # (m-1, m-1) — shared
    vn = v_null if torch.is_tensor(v_null) else torch.tensor(
        v_null, dtype=V.dtype)
    vn = vn.reshape(-1) if vn.dim() else vn.reshape(1)
    span = (v_full - vn)               # (R,)
    Yr = V - vn.reshape(-1, 1) - span.reshape(-1, 1) * zl.unsqueeze(0)
    B = Zw.t() @ Yr.t()                # (m-1, R)
    eye = torch.eye(m - 1, dtype=A.dtype)
    phi_rest = torch.linalg.solve(A + 1e-10 * eye, B).t()  # (R, m-1)
    phi_last = span - phi_rest.sum(dim=1)
    return torch.cat([phi_rest, phi_last.reshape(R, 1)], dim=1)
