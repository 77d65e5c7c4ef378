"""AutoTrainer layer (core/train parity): TrainClassifier / TrainRegressor
(auto-featurize + label index + fit any estimator), ComputeModelStatistics
(confusion matrix / AUC / precision-recall / regression metrics),
ComputePerInstanceStatistics."""
from __future__ import annotations


import numpy as np
import pandas as pd

from ..core.param import Param, toBool, toInt, toList, toString
from ..core.pipeline import Estimator, Model, Transformer
from ..core.registry import register
from .featurize import Featurize


@register
class TrainClassifier(Estimator):
    """Auto-featurize + reindex labels + fit the inner classifier
    (core/.../train/TrainClassifier.scala:49)."""
    model = Param("model", "inner classifier estimator", None, is_complex=True)
    labelCol = Param("labelCol", "label column", "label")
    featuresCol = Param("featuresCol", "generated features column",
                        "TrainClassifier_features")
    numFeatures = Param("numFeatures", "hash dim for high-card strings", 0, toInt)
    reindexLabel = Param("reindexLabel", "re-index labels to 0..K-1 "
                         "(TrainClassifier.scala reindexLabel); False trusts "
                         "numeric labels as-is", True, toBool)

    def _fit(self, df: pd.DataFrame):
        label = self.get("labelCol")
        inner = self.get("model")
        if inner is None:
            from ..models.gbdt.estimators import LightGBMClassifier
            inner = LightGBMClassifier()
        if not self.get("reindexLabel"):
            levels = sorted({str(v) for v in pd.unique(df[label].dropna())})
            lut = {str(v): float(v) for v in pd.unique(df[label].dropna())}
        else:
            levels = sorted((str(v) for v in pd.unique(df[label].dropna())))
            lut = {v: i for i, v in enumerate(levels)}
        feat_cols = [c for c in df.columns if c != label]
        featurizer = Featurize(inputCols=feat_cols,
                               outputCol=self.get("featuresCol")).fit(df)
        dff = featurizer.transform(df)
        dff[label] = [lut[str(v)] for v in df[label]]
        inner = inner.copy()
        inner.set("labelCol", label)
        inner.set("featuresCol", self.get("featuresCol"))
        fitted = inner.fit(dff)
        m = TrainedClassifierModel(featurizer=featurizer, inner=fitted)
        m.set("labelCol", label)
        # without reindexing there is no index→level mapping to invert
        if self.get("reindexLabel"):
            m.set("levels", levels)
        m.set("featuresCol", self.get("featuresCol"))
        return m


@register
class TrainedClassifierModel(Model):
    labelCol = Param("labelCol", "label column", "label")
    levels = Param("levels", "label levels", None, toList)
    featuresCol = Param("featuresCol", "features column", None)
    featurizerModel = Param("featurizerModel", "fitted featurizer", None,
                            is_complex=True)
    innerModel = Param("innerModel", "fitted classifier", None, is_complex=True)

    def __init__(self, featurizer=None, inner=None, **kwargs):
        super().__init__(**kwargs)
        if featurizer is not None:
            self.set("featurizerModel", featurizer)
        if inner is not None:
            self.set("innerModel", inner)

    def _transform(self, df):
        dff = self.get("featurizerModel").transform(df)
        out = self.get("innerModel").transform(dff)
        levels = self.get("levels")
        if levels and "prediction" in out.columns:
            out["scored_labels"] = [
                levels[int(p)] if 0 <= int(p) < len(levels) else None
                for p in out["prediction"]]
        return out


@register
class TrainRegressor(Estimator):
    model = Param("model", "inner regressor estimator", None, is_complex=True)
    labelCol = Param("labelCol", "label column", "label")
    featuresCol = Param("featuresCol", "generated features column",
                        "TrainRegressor_features")

    def _fit(self, df):
        label = self.get("labelCol")
        inner = self.get("model")
        if inner is None:
            from ..models.gbdt.estimators import LightGBMRegressor
            inner = LightGBMRegressor()
        feat_cols = [c for c in df.columns if c != label]
        featurizer = Featurize(inputCols=feat_cols,
                               outputCol=self.get("featuresCol")).fit(df)
        dff = featurizer.transform(df)
        inner = inner.copy()
        inner.set("labelCol", label)
        inner.set("featuresCol", self.get("featuresCol"))
        fitted = inner.fit(dff)
        m = TrainedRegressorModel(featurizer=featurizer, inner=fitted)
        m.set("labelCol", label)
        return m


@register
class TrainedRegressorModel(Model):
    labelCol = Param("labelCol", "label column", "label")
    featurizerModel = Param("featurizerModel", "fitted featurizer", None,
                            is_complex=True)
    innerModel = Param("innerModel", "fitted regressor", None, is_complex=True)

    def __init__(self, featurizer=None, inner=None, **kwargs):
        super().__init__(**kwargs)
        if featurizer is not None:
            self.set("featurizerModel", featurizer)
        if inner is not None:
            self.set("innerModel", inner)

    def _transform(self, df):
        return self.get("innerModel").transform(
            self.get("featurizerModel").transform(df))


@register
class ComputeModelStatistics(Transformer):
    """Classification/regression metrics DataFrame
    (train/ComputeModelStatistics.scala:58; multiclass formulas :330-371)."""
    labelCol = Param("labelCol", "label column", "label")
    scoresCol = Param("scoresCol", "probability column", "probability")
    scoredLabelsCol = Param("scoredLabelsCol", "prediction column", "prediction")
    evaluationMetric = Param("evaluationMetric", "classification|regression|"
                             "all", "all", toString)

    def _transform(self, df: pd.DataFrame) -> pd.DataFrame:
        y = df[self.get("labelCol")].to_numpy()
        kind = self.get("evaluationMetric")
        is_classification = kind == "classification" or (
            kind == "all" and self.get("scoredLabelsCol") in df.columns
            and df[self.get("scoredLabelsCol")].nunique() <= max(20, int(
                np.sqrt(len(df)))) and not np.issubdtype(y.dtype, np.floating)
        ) or (kind == "all" and set(np.unique(y.astype(float))) <= set(
            np.arange(100).astype(float)))
        if is_classification:
            yp = df[self.get("scoredLabelsCol")].to_numpy().astype(float)
            ya = y.astype(float)
            classes = np.unique(np.concatenate([ya, yp]))
            k = len(classes)
            lut = {c: i for i, c in enumerate(classes)}
            cm = np.zeros((k, k), dtype=np.int64)
            for a, p in zip(ya, yp):
                cm[lut[a], lut[p]] += 1
            acc = float(np.trace(cm)) / max(cm.sum(), 1)
            prec = np.diag(cm) / np.maximum(cm.sum(axis=0), 1)
            rec = np.diag(cm) / np.maximum(cm.sum(axis=1), 1)
            row = {"accuracy": acc,
                   "precision": float(np.average(prec, weights=cm.sum(axis=1))),
                   "recall": float(np.average(rec, weights=cm.sum(axis=1))),
                   "confusion_matrix": cm.tolist()}
            if k == 2 and self.get("scoresCol") in df.columns:
                from ..models.gbdt.metrics import auc as _auc
                scores = df[self.get("scoresCol")].to_numpy()
                p1 = np.stack([np.asarray(v) for v in scores])[:, 1] \
                    if isinstance(scores[0], (list, np.ndarray)) else scores
                row["AUC"] = _auc(np.asarray(p1, dtype=np.float64), ya)
            return pd.DataFrame([row])
        # regression
        yp = df[self.get("scoredLabelsCol")].to_numpy().astype(float)
        ya = y.astype(float)
        mse = float(np.mean((yp - ya) ** 2))
        var = float(np.var(ya))
        return pd.DataFrame([{
            "mean_squared_error": mse,
            "root_mean_squared_error": float(np.sqrt(mse)),
            "mean_absolute_error": float(np.mean(np.abs(yp - ya))),
            "R^2": 1.0 - mse / max(var, 1e-12),
        }])


@register
class ComputePerInstanceStatistics(Transformer):
    """Per-row loss/error columns (ComputePerInstanceStatistics.scala)."""
    labelCol = Param("labelCol", "label column", "label")
    scoresCol = Param("scoresCol", "probability column", "probability")
    scoredLabelsCol = Param("scoredLabelsCol", "prediction column", "prediction")

    def _transform(self, df):
        out = df.copy()
        y = df[self.get("labelCol")].to_numpy().astype(float)
        yp = df[self.get("scoredLabelsCol")].to_numpy().astype(float)
        if self.get("scoresCol") in df.columns and len(df) and isinstance(
                df[self.get("scoresCol")].iloc[0], (list, np.ndarray)):
            probs = np.stack([np.asarray(v) for v in df[self.get("scoresCol")]])
            eps = 1e-15
            ll = -np.log(np.clip(
                probs[np.arange(len(df)), y.astype(int)], eps, 1.0))
            out["log_loss"] = ll
        out["L1_loss"] = np.abs(yp - y)
        out["L2_loss"] = (yp - y) ** 2
        return out
