"""KNN / ConditionalKNN — exact nearest-neighbor search, MI355X-first.

The reference broadcasts a ball tree and runs a bound-pruned DFS per row UDF
(core/.../nn/BallTree.scala:109, ConditionalKNN.scala:32-100).  On MI355X the
idiomatic design is brute-force tiled distance computation on the matrix
cores (torch matmul → rocBLAS MFMA GEMM: ||q-x||² = ||q||² - 2q·x + ||x||²)
+ device top-k, batched over queries — exact, and faster than tree traversal
on 8 TB/s HBM for the data sizes the reference targets.  ConditionalKNN
masks disallowed labels before the top-k (label-filtered search parity)."""
from __future__ import annotations

from typing import List, Optional

import numpy as np
import pandas as pd
import torch

from ..core.param import Param, toInt
from ..core.pipeline import Estimator, Model
from ..core.registry import register
from ..core.schema import features_matrix
from ..utils.devices import default_device


class _KNNParamsMixin:
    featuresCol = Param("featuresCol", "query features column", "features")
    valuesCol = Param("valuesCol", "payload column carried to matches", "values")
    outputCol = Param("outputCol", "matches output column", "output")
    k = Param("k", "number of neighbors", 5, toInt)
    batchSize = Param("batchSize", "query rows per device batch", 4096, toInt)
    leafSize = Param("leafSize", "ball-tree leaf size (used only by the CPU "
                     "BallTree view, getBallTree(); the GPU path is exact "
                     "brute-force — KNN.scala leafSize)", 50, toInt)


@register
class KNN(_KNNParamsMixin, Estimator):
    """Fit = index the reference DataFrame (KNN.scala:48)."""

    def _fit(self, df: pd.DataFrame):
        X = features_matrix(df, self.get("featuresCol"), None)
        values = df[self.get("valuesCol")].tolist() \
            if self.get("valuesCol") in df.columns else list(range(len(df)))
        model = KNNModel(index=X, values=values)
        for p in ("featuresCol", "outputCol", "k", "batchSize", "leafSize"):
            model.set(p, self.get(p))
        return model


@register
class KNNModel(_KNNParamsMixin, Model):
    indexData = Param("indexData", "indexed matrix + payload", None,
                      is_complex=True)

    def getBallTree(self):
        """The reference's broadcast ball tree (BallTree.scala:109) over the
        same index; built lazily on the CPU.  Scoring here uses the exact
        brute-force MFMA path, so this exists for API/introspection parity
        and CPU-side traversal."""
        from .balltree import BallTree
        if getattr(self, "_ball_tree", None) is None:
            d = self.get("indexData")
            self._ball_tree = BallTree(d["index"], list(d["values"]),
                                       leaf_size=self.get("leafSize"))
        return self._ball_tree

    def __init__(self, index: Optional[np.ndarray] = None,
                 values: Optional[List] = None, labels=None, **kwargs):
        super().__init__(**kwargs)
        if index is not None:
            payload = {"index": np.ascontiguousarray(index, dtype=np.float32)}
            payload["values"] = np.asarray(values, dtype=object).astype(str) \
                if values is not None and not np.issubdtype(
                    np.asarray(values).dtype, np.number) \
                else np.asarray(values if values is not None else [])
            if labels is not None:
                payload["labels"] = np.asarray(labels)
            self.set("indexData", {k: v for k, v in payload.items()})

    def _search(self, Q: np.ndarray, cond_masks=None, cond_bits=None,
                label_ids=None):
        """Tiled MFMA-GEMM distances + device top-k.

        Conditioning: either host bool masks (small corpora), or — the
        large-corpus path — per-query uint64 label bitmasks tested on
        device against the index's label-id vector, so no (n_q, n_index)
        host matrix ever exists."""
        data = self.get("indexData")
        device = default_device("auto")
        X = torch.from_numpy(data["index"]).to(device)
        xsq = (X * X).sum(dim=1)
        k = min(self.get("k"), X.shape[0])
        # memory-aware batch clamp: the (bs, n) distance tile (plus the
        # transient mask tile) must stay bounded as the corpus grows
        bs = min(self.get("batchSize"),
                 max(128, int((1 << 31) / max(1, X.shape[0]))))
        lid = (torch.from_numpy(label_ids).to(device)
               if label_ids is not None else None)
        all_idx, all_dist = [], []
        for s in range(0, len(Q), bs):
            q = torch.from_numpy(Q[s:s + bs]).to(device)
            d2 = (q * q).sum(1, keepdim=True) - 2.0 * (q @ X.t()) + xsq
            if cond_bits is not None:
                bits = torch.from_numpy(cond_bits[s:s + bs]).to(device)
                allowed = ((bits.unsqueeze(1) >> lid.unsqueeze(0)) & 1).bool()
                d2 = torch.where(allowed, d2,
                                 torch.full_like(d2, float("inf")))
            elif cond_masks is not None:
                m = torch.from_numpy(cond_masks[s:s + bs]).to(device)
                d2 = torch.where(m, d2, torch.full_like(d2, float("inf")))
            dist, idx = torch.topk(d2, k, dim=1, largest=False)
            all_idx.append(idx.cpu().numpy())
            all_dist.append(dist.clamp_min(0).sqrt().cpu().numpy())
        return np.concatenate(all_idx), np.concatenate(all_dist)

    def _transform(self, df: pd.DataFrame) -> pd.DataFrame:
        Q = features_matrix(df, self.get("featuresCol"), None)
        data = self.get("indexData")
        vals = data.get("values")
        idx, dist = self._search(Q) if len(df) else (np.zeros((0, 0), int),
                                                     np.zeros((0, 0)))
        has_vals = len(vals) > 0
        matches = [
            [{"value": (vals[j].item() if hasattr(vals[j], "item")
                        else vals[j]) if has_vals else j,
              "distance": d}
             for j, d in zip(ir, dr)]
            for ir, dr in zip(idx.tolist(), dist.tolist())]
        out = df.copy()
        out[self.get("outputCol")] = matches
        return out


@register
class ConditionalKNN(_KNNParamsMixin, Estimator):
    """Label-conditioned KNN (ConditionalKNN.scala:32): each query carries a
    set of allowed labels; only index points with those labels match."""
    labelCol = Param("labelCol", "index label column", "labels")
    conditionerCol = Param("conditionerCol", "query allowed-label-set column",
                           "conditioner")

    def _fit(self, df: pd.DataFrame):
        X = features_matrix(df, self.get("featuresCol"), None)
        values = df[self.get("valuesCol")].tolist() \
            if self.get("valuesCol") in df.columns else list(range(len(df)))
        labels = df[self.get("labelCol")].to_numpy()
        model = ConditionalKNNModel(index=X, values=values, labels=labels)
        for p in ("featuresCol", "outputCol", "k", "batchSize",
                  "conditionerCol"):
            model.set(p, self.get(p))
        return model


@register
class ConditionalKNNModel(KNNModel):
    conditionerCol = Param("conditionerCol", "query allowed-label-set column",
                           "conditioner")

    def _transform(self, df: pd.DataFrame) -> pd.DataFrame:
        Q = features_matrix(df, self.get("featuresCol"), None)
        data = self.get("indexData")
        labels = data["labels"]
        vals = data.get("values")
        conds = df[self.get("conditionerCol")].to_numpy()
        uniq, label_ids = np.unique(labels, return_inverse=True)
        if len(df) == 0:
            idx, dist = np.zeros((0, 0), int), np.zeros((0, 0))
        elif len(uniq) <= 64:
            # large-corpus path: per-query uint64 allowed-label bitmask,
            # tested on device — no (n_q, n_index) host matrix
            pos = {v: i for i, v in enumerate(uniq)}
            bits = np.zeros(len(df), dtype=np.int64)
            for i, c in enumerate(conds):
                allowed = (c if isinstance(c, (list, tuple, set, np.ndarray))
                           else [c])
                b = 0
                for v in allowed:
                    j = pos.get(v)
                    if j is not None:
                        b |= (1 << j)
                bits[i] = b
            idx, dist = self._search(Q, cond_bits=bits,
                                     label_ids=label_ids.astype(np.int64))
        else:
            masks = np.zeros((len(df), len(labels)), dtype=bool)
            for i, c in enumerate(conds):
                allowed = set(c) if isinstance(
                    c, (list, tuple, set, np.ndarray)) else {c}
                masks[i] = np.isin(labels, list(allowed))
            idx, dist = self._search(Q, masks)
        matches = []
        for i in range(len(df)):
            row = []
            for c, j in enumerate(idx[i]):
                if np.isfinite(dist[i, c]):
                    row.append({
                        "value": (vals[j].item() if hasattr(vals[j], "item")
                                  else vals[j]) if len(vals) else int(j),
                        "distance": float(dist[i, c]),
                        "label": labels[j].item() if hasattr(labels[j], "item")
                                 else labels[j]})
            matches.append(row)
        out = df.copy()
        out[self.get("outputCol")] = matches
        return out
