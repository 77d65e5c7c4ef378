"""CyberML — anomalous-access detection (core/src/main/python/mmlspark/cyber
parity: anomaly/collaborative_filtering.py AccessAnomaly over ALS,
complement_access.py ComplementAccessTransformer, feature/indexers.py
IdIndexer, feature/scalers.py StandardScalarScaler/LinearScalarScaler).

The ALS factorization runs in torch (normal equations per user/resource —
dense solves batch on the GPU), per-tenant.  Scores are user·resource
affinities; anomalous = low affinity, standardized per tenant so the output
is in z-score units (higher = more anomalous)."""
from __future__ import annotations

from typing import Dict

import numpy as np
import pandas as pd
import torch

from ..core.param import Param, toFloat, toInt
from ..core.pipeline import Estimator, Model, Transformer
from ..core.registry import register


@register
class ComplementAccessTransformer(Transformer):
    """Sample (user, resource) pairs NOT present in the access log
    (complement_access.py:148)."""
    tenantCol = Param("tenantCol", "tenant column", "tenant")
    indexedUserCol = Param("indexedUserCol", "user index column", "user_idx")
    indexedResCol = Param("indexedResCol", "resource index column", "res_idx")
    complementsetFactor = Param("complementsetFactor",
                                "complement rows per observed row", 2, toInt)
    seed = Param("seed", "sampling seed", 0, toInt)

    def _transform(self, df: pd.DataFrame) -> pd.DataFrame:
        rng = np.random.default_rng(self.get("seed"))
        tcol, ucol, rcol = (self.get("tenantCol"), self.get("indexedUserCol"),
                            self.get("indexedResCol"))
        out_rows = []
        for tenant, g in df.groupby(tcol, sort=False):
            seen = set(zip(g[ucol], g[rcol]))
            users = g[ucol].unique()
            ress = g[rcol].unique()
            want = len(g) * self.get("complementsetFactor")
            tries = 0
            got = 0
            while got < want and tries < want * 20:
                u = users[rng.integers(0, len(users))]
                r = ress[rng.integers(0, len(ress))]
                tries += 1
                if (u, r) not in seen:
                    seen.add((u, r))
                    out_rows.append({tcol: tenant, ucol: u, rcol: r})
                    got += 1
        return pd.DataFrame(out_rows, columns=[tcol, ucol, rcol])


@register
class IdIndexer(Estimator):
    """Per-tenant contiguous id indexing (feature/indexers.py:136)."""
    inputCol = Param("inputCol", "raw id column", None)
    partitionKey = Param("partitionKey", "tenant column", "tenant")
    outputCol = Param("outputCol", "indexed column", None)
    resetPerPartition = Param("resetPerPartition", "restart ids per tenant",
                              True)

    def _fit(self, df):
        maps: Dict = {}
        if self.get("resetPerPartition"):
            for tenant, g in df.groupby(self.get("partitionKey"), sort=False):
                maps[str(tenant)] = {str(v): i for i, v in
                                     enumerate(pd.unique(g[self.get("inputCol")]))}
        else:
            maps["__all__"] = {str(v): i for i, v in
                               enumerate(pd.unique(df[self.get("inputCol")]))}
        m = IdIndexerModel()
        m.set("idMaps", maps)
        for p in ("inputCol", "partitionKey", "outputCol", "resetPerPartition"):
            m.set(p, self.get(p))
        return m


@register
class IdIndexerModel(Model):
    inputCol = Param("inputCol", "raw id column", None)
    partitionKey = Param("partitionKey", "tenant column", "tenant")
    outputCol = Param("outputCol", "indexed column", None)
    resetPerPartition = Param("resetPerPartition", "restart per tenant", True)
    idMaps = Param("idMaps", "tenant → id map", None, is_complex=True)

    def _transform(self, df):
        maps = self.get("idMaps")
        out = df.copy()
        if self.get("resetPerPartition"):
            out[self.get("outputCol")] = [
                maps.get(str(t), {}).get(str(v), -1)
                for t, v in zip(df[self.get("partitionKey")],
                                df[self.get("inputCol")])]
        else:
            m = maps["__all__"]
            out[self.get("outputCol")] = [m.get(str(v), -1)
                                          for v in df[self.get("inputCol")]]
        return out


@register
class StandardScalarScaler(Estimator):
    """Per-tenant z-score scaler (feature/scalers.py:325)."""
    inputCol = Param("inputCol", "value column", None)
    partitionKey = Param("partitionKey", "tenant column", "tenant")
    outputCol = Param("outputCol", "scaled column", None)
    coefficientFactor = Param("coefficientFactor", "output multiplier", 1.0,
                              toFloat)

    def _fit(self, df):
        stats = {}
        for tenant, g in df.groupby(self.get("partitionKey"), sort=False):
            v = g[self.get("inputCol")].to_numpy(dtype=np.float64)
            stats[str(tenant)] = [float(v.mean()), float(v.std() + 1e-12)]
        m = StandardScalarScalerModel()
        m.set("stats", stats)
        for p in ("inputCol", "partitionKey", "outputCol", "coefficientFactor"):
            m.set(p, self.get(p))
        return m


@register
class StandardScalarScalerModel(Model):
    inputCol = Param("inputCol", "value column", None)
    partitionKey = Param("partitionKey", "tenant column", "tenant")
    outputCol = Param("outputCol", "scaled column", None)
    coefficientFactor = Param("coefficientFactor", "multiplier", 1.0, toFloat)
    stats = Param("stats", "tenant → (mean, std)", None, is_complex=True)

    def _transform(self, df):
        stats = self.get("stats")
        c = self.get("coefficientFactor")
        out = df.copy()
        vals = []
        for t, v in zip(df[self.get("partitionKey")],
                        df[self.get("inputCol")]):
            mu, sd = stats.get(str(t), [0.0, 1.0])
            vals.append(c * (float(v) - mu) / sd)
        out[self.get("outputCol")] = vals
        return out


@register
class LinearScalarScaler(Estimator):
    """Per-tenant min-max → [minRequiredValue, maxRequiredValue]."""
    inputCol = Param("inputCol", "value column", None)
    partitionKey = Param("partitionKey", "tenant column", "tenant")
    outputCol = Param("outputCol", "scaled column", None)
    minRequiredValue = Param("minRequiredValue", "output min", 0.0, toFloat)
    maxRequiredValue = Param("maxRequiredValue", "output max", 1.0, toFloat)

    def _fit(self, df):
        stats = {}
        for tenant, g in df.groupby(self.get("partitionKey"), sort=False):
            v = g[self.get("inputCol")].to_numpy(dtype=np.float64)
            stats[str(tenant)] = [float(v.min()), float(v.max())]
        m = LinearScalarScalerModel()
        m.set("stats", stats)
        for p in ("inputCol", "partitionKey", "outputCol", "minRequiredValue",
                  "maxRequiredValue"):
            m.set(p, self.get(p))
        return m


@register
class LinearScalarScalerModel(Model):
    inputCol = Param("inputCol", "value column", None)
    partitionKey = Param("partitionKey", "tenant column", "tenant")
    outputCol = Param("outputCol", "scaled column", None)
    minRequiredValue = Param("minRequiredValue", "output min", 0.0, toFloat)
    maxRequiredValue = Param("maxRequiredValue", "output max", 1.0, toFloat)
    stats = Param("stats", "tenant → (min, max)", None, is_complex=True)

    def _transform(self, df):
        stats = self.get("stats")
        lo, hi = self.get("minRequiredValue"), self.get("maxRequiredValue")
        out = df.copy()
        vals = []
        for t, v in zip(df[self.get("partitionKey")], df[self.get("inputCol")]):
            mn, mx = stats.get(str(t), [0.0, 1.0])
            scale = (hi - lo) / max(mx - mn, 1e-12)
            vals.append(lo + (float(v) - mn) * scale)
        out[self.get("outputCol")] = vals
        return out


def _als(n_u: int, n_r: int, u_idx, r_idx, ratings, rank=10, reg=0.1,
         iters=10, device="cpu", seed=0):
    """Regularized ALS on sparse triples via batched normal equations."""
    g = torch.Generator().manual_seed(seed)
    U = torch.randn(n_u, rank, generator=g).to(device) * 0.1
    V = torch.randn(n_r, rank, generator=g).to(device) * 0.1
    u = torch.as_tensor(u_idx, dtype=torch.long, device=device)
    r = torch.as_tensor(r_idx, dtype=torch.long, device=device)
    y = torch.as_tensor(ratings, dtype=torch.float32, device=device)
    eye = torch.eye(rank, device=device)

    def solve_side(fix, fix_idx, var_idx, n_var):
        F = fix[fix_idx]  # (nnz, k)
        # per-var Gram matrices and rhs via index_add
        G = torch.zeros(n_var, rank, rank, device=device)
        b = torch.zeros(n_var, rank, device=device)
        G.index_add_(0, var_idx, F.unsqueeze(2) * F.unsqueeze(1))
        b.index_add_(0, var_idx, F * y.unsqueeze(1))
        G = G + reg * eye
        return torch.linalg.solve(G, b)

    for _ in range(iters):
        U = solve_side(V, r, u, n_u)
        V = solve_side(U, u, r, n_r)
    return U, V


@register
class AccessAnomaly(Estimator):
    """Collaborative-filtering access-anomaly estimator
    (cyber/anomaly/collaborative_filtering.py AccessAnomaly)."""
    tenantCol = Param("tenantCol", "tenant column", "tenant")
    indexedUserCol = Param("indexedUserCol", "user index column", "user_idx")
    indexedResCol = Param("indexedResCol", "resource index column", "res_idx")
    rankParam = Param("rankParam", "ALS rank", 10, toInt)
    regParam = Param("regParam", "ALS regularization", 0.1, toFloat)
    maxIter = Param("maxIter", "ALS iterations", 10, toInt)
    complementsetFactor = Param("complementsetFactor",
                                "negative samples per observed", 2, toInt)
    negScore = Param("negScore", "rating for complement samples", 0.0, toFloat)
    outputCol = Param("outputCol", "anomaly score column", "anomaly_score")
    seed = Param("seed", "seed", 0, toInt)

    def _fit(self, df: pd.DataFrame):
        tcol = self.get("tenantCol")
        ucol, rcol = self.get("indexedUserCol"), self.get("indexedResCol")
        comp = ComplementAccessTransformer(
            tenantCol=tcol, indexedUserCol=ucol, indexedResCol=rcol,
            complementsetFactor=self.get("complementsetFactor"),
            seed=self.get("seed"))
        factors = {}
        train_scores = []
        for tenant, g in df.groupby(tcol, sort=False):
            neg = comp.transform(g)
            n_u = int(g[ucol].max()) + 1
            n_r = int(g[rcol].max()) + 1
            u_idx = np.concatenate([g[ucol].to_numpy(),
                                    neg[ucol].to_numpy()]) if len(neg) \
                else g[ucol].to_numpy()
            r_idx = np.concatenate([g[rcol].to_numpy(),
                                    neg[rcol].to_numpy()]) if len(neg) \
                else g[rcol].to_numpy()
            ratings = np.concatenate([
                np.ones(len(g)),
                np.full(len(neg), self.get("negScore"))]) if len(neg) \
                else np.ones(len(g))
            U, V = _als(n_u, n_r, u_idx, r_idx, ratings,
                        rank=self.get("rankParam"), reg=self.get("regParam"),
                        iters=self.get("maxIter"), seed=self.get("seed"))
            factors[str(tenant)] = (U.cpu().numpy(), V.cpu().numpy())
            s = (U[g[ucol].to_numpy()] * V[g[rcol].to_numpy()]).sum(dim=1)
            train_scores.append(pd.DataFrame({
                tcol: tenant, "__raw": (-s).cpu().numpy()}))
        # per-tenant standardization of the anomaly direction (-affinity)
        scaler = StandardScalarScaler(
            inputCol="__raw", partitionKey=tcol, outputCol="__scaled").fit(
                pd.concat(train_scores, ignore_index=True))
        model = AccessAnomalyModel()
        model.set("userFactors", {k: f[0] for k, f in factors.items()})
        model.set("resFactors", {k: f[1] for k, f in factors.items()})
        model.set("scalerStats", scaler.get("stats"))
        for p in ("tenantCol", "indexedUserCol", "indexedResCol", "outputCol"):
            model.set(p, self.get(p))
        return model


@register
class AccessAnomalyModel(Model):
    tenantCol = Param("tenantCol", "tenant column", "tenant")
    indexedUserCol = Param("indexedUserCol", "user index column", "user_idx")
    indexedResCol = Param("indexedResCol", "resource index column", "res_idx")
    outputCol = Param("outputCol", "anomaly score column", "anomaly_score")
    userFactors = Param("userFactors", "tenant → U", None, is_complex=True)
    resFactors = Param("resFactors", "tenant → V", None, is_complex=True)
    scalerStats = Param("scalerStats", "tenant → (mean,std)", None,
                        is_complex=True)

    def _transform(self, df: pd.DataFrame) -> pd.DataFrame:
        tcol = self.get("tenantCol")
        ucol, rcol = self.get("indexedUserCol"), self.get("indexedResCol")
        uf, vf = self.get("userFactors"), self.get("resFactors")
        stats = self.get("scalerStats")
        scores = []
        for t, u, r in zip(df[tcol], df[ucol], df[rcol]):
            U = uf.get(str(t))
            V = vf.get(str(t))
            if U is None or not (0 <= u < len(U)) or not (0 <= r < len(V)):
                scores.append(float("nan"))
                continue
            raw = -float((U[int(u)] * V[int(r)]).sum())
            mu, sd = stats.get(str(t), [0.0, 1.0])
            scores.append((raw - mu) / sd)
        out = df.copy()
        out[self.get("outputCol")] = scores
        return out
