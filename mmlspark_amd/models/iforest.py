"""Isolation Forest — native implementation (the reference only re-wraps
LinkedIn's Scala isolation-forest: core/.../isolationforest/IsolationForest.scala:18-63;
here the subsampled random-tree build and path-length scoring are ours).

Scoring is array-based tree traversal (torch, CPU/GPU-generic); the anomaly
score is 2^(-E[h(x)]/c(n)) with the standard average-path normalizer."""
from __future__ import annotations

import math
from typing import List, Optional

import numpy as np
import pandas as pd
import torch

from ..core.param import Param, toBool, toFloat, toInt
from ..core.pipeline import Estimator, Model
from ..core.registry import register
from ..core.schema import features_matrix
from ..utils.devices import default_device


def _avg_path(n: float) -> float:
    if n <= 1:
        return 0.0
    return 2.0 * (math.log(n - 1) + 0.5772156649) - 2.0 * (n - 1) / n


class _ITree:
    __slots__ = ("feature", "threshold", "left", "right", "size")

    def __init__(self):
        self.feature: List[int] = []
        self.threshold: List[float] = []
        self.left: List[int] = []
        self.right: List[int] = []
        self.size: List[int] = []

    def new_node(self):
        self.feature.append(-1)
        self.threshold.append(0.0)
        self.left.append(-1)
        self.right.append(-1)
        self.size.append(0)
        return len(self.feature) - 1


def _build_tree(X: np.ndarray, rng: np.random.Generator, height_limit: int,
                feat_pool: np.ndarray) -> _ITree:
    tree = _ITree()

    def rec(rows: np.ndarray, depth: int) -> int:
        nid = tree.new_node()
        tree.size[nid] = len(rows)
        if depth >= height_limit or len(rows) <= 1:
            return nid
        sub = X[rows]
        for _ in range(8):  # retry constant features
            f = int(rng.choice(feat_pool))
            lo, hi = sub[:, f].min(), sub[:, f].max()
            if hi > lo:
                break
        else:
            return nid
        thr = float(rng.uniform(lo, hi))
        mask = sub[:, f] < thr
        tree.feature[nid] = f
        tree.threshold[nid] = thr
        tree.left[nid] = rec(rows[mask], depth + 1)
        tree.right[nid] = rec(rows[~mask], depth + 1)
        return nid

    rec(np.arange(len(X)), 0)
    return tree


@register
class IsolationForest(Estimator):
    featuresCol = Param("featuresCol", "features column", "features")
    featureCols = Param("featureCols", "numeric feature columns", None)
    predictionCol = Param("predictionCol", "0/1 outlier label column",
                          "predictedLabel")
    scoreCol = Param("scoreCol", "anomaly score column", "outlierScore")
    numEstimators = Param("numEstimators", "number of trees", 100, toInt)
    maxSamples = Param("maxSamples", "subsample size per tree", 256, toFloat)
    maxFeatures = Param("maxFeatures", "feature fraction per tree", 1.0, toFloat)
    bootstrap = Param("bootstrap", "sample with replacement", False, toBool)
    contamination = Param("contamination", "expected outlier fraction (0 = "
                          "use score threshold 0.5)", 0.0, toFloat)
    randomSeed = Param("randomSeed", "seed", 1, toInt)

    def _fit(self, df: pd.DataFrame):
        X = features_matrix(df, self.get("featuresCol"), self.get("featureCols"))
        rng = np.random.default_rng(self.get("randomSeed"))
        n, nf = X.shape
        ms = self.get("maxSamples")
        sub_n = int(ms if ms > 1 else max(2, ms * n))
        sub_n = min(sub_n, n)
        height = math.ceil(math.log2(max(sub_n, 2)))
        n_feat = max(1, int(self.get("maxFeatures") * nf))
        trees = []
        for _ in range(self.get("numEstimators")):
            rows = (rng.integers(0, n, size=sub_n) if self.get("bootstrap")
                    else rng.permutation(n)[:sub_n])
            feat_pool = rng.permutation(nf)[:n_feat]
            trees.append(_build_tree(X[rows], rng, height, feat_pool))
        model = IsolationForestModel(trees=trees, sub_n=sub_n)
        for p in ("featuresCol", "featureCols", "predictionCol", "scoreCol"):
            model.set(p, self.get(p))
        # contamination → score threshold on train scores
        if self.get("contamination") > 0:
            scores = model._scores(X)
            thr = float(np.quantile(scores, 1 - self.get("contamination")))
            model.set("scoreThreshold", thr)
        return model


@register
class IsolationForestModel(Model):
    featuresCol = Param("featuresCol", "features column", "features")
    featureCols = Param("featureCols", "numeric feature columns", None)
    predictionCol = Param("predictionCol", "0/1 outlier label", "predictedLabel")
    scoreCol = Param("scoreCol", "anomaly score column", "outlierScore")
    scoreThreshold = Param("scoreThreshold", "outlier score cutoff", 0.5, toFloat)
    forestArrays = Param("forestArrays", "flattened forest", None, is_complex=True)

    def __init__(self, trees: Optional[List[_ITree]] = None,
                 sub_n: int = 256, **kwargs):
        super().__init__(**kwargs)
        if trees is not None:
            offs = [0]
            feat, thr, lft, rgt, size = [], [], [], [], []
            for t in trees:
                feat.extend(t.feature)
                thr.extend(t.threshold)
                lft.extend(t.left)
                rgt.extend(t.right)
                size.extend(t.size)
                offs.append(len(feat))
            self.set("forestArrays", {
                "feature": np.asarray(feat, np.int32),
                "threshold": np.asarray(thr, np.float32),
                "left": np.asarray(lft, np.int32),
                "right": np.asarray(rgt, np.int32),
                "size": np.asarray(size, np.float32),
                "offsets": np.asarray(offs, np.int64),
                "sub_n": np.asarray([sub_n], np.int64),
            })

    def _leaf_depth_values(self, f):
        """Per-node value = depth + avg_path(leaf size) for leaves (0 for
        internal nodes): the whole path-length estimate becomes ONE forest
        traversal through the batched HIP predict kernel."""
        feat, lft, rgt = f["feature"], f["left"], f["right"]
        size, offs = f["size"], f["offsets"]
        val = np.zeros(len(feat), dtype=np.float32)
        for t in range(len(offs) - 1):
            base, end = int(offs[t]), int(offs[t + 1])
            depth = np.zeros(end - base, dtype=np.float32)
            for i in range(base, end):  # parents precede children
                if feat[i] >= 0:
                    depth[lft[i] ] = depth[i - base] + 1
                    depth[rgt[i] ] = depth[i - base] + 1
            for i in range(base, end):
                if feat[i] < 0:
                    val[i] = depth[i - base] + _avg_path(
                        max(float(size[i]), 1.0))
        return val

    def _scores(self, X: np.ndarray) -> np.ndarray:
        from ..ops import backend
        f = self.get("forestArrays")
        if "leaf_value" not in f:
            f = dict(f)
            f["leaf_value"] = self._leaf_depth_values(f)
            self.set("forestArrays", f)
        device = default_device("auto")
        sub_n = int(f["sub_n"][0])
        n_trees = len(f["offsets"]) - 1
        Xt = torch.from_numpy(np.ascontiguousarray(X)).to(device)
        dev = lambda a, dt: torch.from_numpy(  # noqa: E731
            np.ascontiguousarray(a)).to(device).to(dt)
        # shift thresholds so the kernel's `x <= thr` matches iforest's
        # `x < thr` (thresholds are continuous uniforms: exact ties are
        # measure-zero; nextafter keeps even those consistent)
        thr = np.nextafter(f["threshold"], -np.inf).astype(np.float32)
        depth_sum = backend.predict_forest(
            dev(f["feature"], torch.int32), dev(thr, torch.float32),
            dev(f["left"], torch.int32), dev(f["right"], torch.int32),
            dev(f["leaf_value"], torch.float32),
            torch.from_numpy(f["offsets"]).to(device), Xt.float(), 1,
            torch.ones(n_trees, device=device))
        avg_depth = depth_sum.reshape(-1) / n_trees
        c = _avg_path(sub_n)
        return (2.0 ** (-avg_depth.cpu().numpy()
                        / max(c, 1e-9))).astype(np.float64)

    def _transform(self, df: pd.DataFrame) -> pd.DataFrame:
        X = features_matrix(df, self.get("featuresCol"), self.get("featureCols"))
        scores = self._scores(X) if len(df) else np.zeros(0)
        out = df.copy()
        out[self.get("scoreCol")] = scores
        out[self.get("predictionCol")] = (
            scores > self.get("scoreThreshold")).astype(np.float64)
        return out
