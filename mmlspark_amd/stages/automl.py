"""AutoML (core/automl parity): HyperparamBuilder/ParamSpace random+grid
search, TuneHyperparameters (parallel CV search, TuneHyperparameters.scala:36),
FindBestModel/BestModel (FindBestModel.scala:50,134)."""
from __future__ import annotations

import concurrent.futures as cf

import numpy as np
import pandas as pd

from ..core.param import Param, toInt, toString
from ..core.pipeline import Estimator, Model
from ..core.registry import register
from .train import ComputeModelStatistics


class DiscreteHyperParam:
    def __init__(self, values):
        self.values = list(values)

    def sample(self, rng):
        return self.values[int(rng.integers(0, len(self.values)))]


class RangeHyperParam:
    def __init__(self, lo, hi, is_int=None, log=False):
        if is_int is None:  # Int vs DoubleRangeHyperParam inferred from bounds
            is_int = isinstance(lo, int) and isinstance(hi, int)
        self.lo, self.hi, self.is_int, self.log = lo, hi, is_int, log

    def sample(self, rng):
        if self.log:
            v = float(np.exp(rng.uniform(np.log(self.lo), np.log(self.hi))))
        else:
            v = float(rng.uniform(self.lo, self.hi))
        return int(round(v)) if self.is_int else v


class HyperparamBuilder:
    """ParamSpace builder (automl/ParamSpace.scala)."""

    def __init__(self):
        self.space = {}

    def addHyperparam(self, name: str, dist):
        self.space[name] = dist
        return self

    def addRange(self, name: str, lo, hi):
        """Convenience for Int/DoubleRangeHyperParam (DefaultHyperparams
        usage, automl/DefaultHyperparams.scala:20-34)."""
        return self.addHyperparam(name, RangeHyperParam(lo, hi))

    def addDiscrete(self, name: str, values):
        return self.addHyperparam(name, DiscreteHyperParam(values))

    def build(self):
        return self.space


def _eval_model(model, df_val, label_col, metric, is_classification):
    out = model.transform(df_val)
    stats = ComputeModelStatistics(
        labelCol=label_col,
        evaluationMetric="classification" if is_classification else "regression"
    ).transform(out)
    row = stats.iloc[0]
    if metric in row:
        return float(row[metric])
    return float(row.get("AUC", row.get("accuracy",
                                        -row.get("mean_squared_error", 0.0))))


@register
class TuneHyperparameters(Estimator):
    """Random search over a param space with K-fold CV, evaluated in a thread
    pool (the reference runs parallel Spark jobs; TuneHyperparameters.fit:144)."""
    evaluationMetric = Param("evaluationMetric", "metric name", "AUC", toString)
    numFolds = Param("numFolds", "CV folds", 3, toInt)
    numRuns = Param("numRuns", "sampled configurations (random mode)", 8, toInt)
    searchMode = Param("searchMode", "random|grid (grid enumerates the "
                       "cartesian product of Discrete params)", "random",
                       toString)
    parallelism = Param("parallelism", "concurrent fits", 4, toInt)
    seed = Param("seed", "sampling seed", 0, toInt)
    labelCol = Param("labelCol", "label column", "label")
    models = Param("models", "candidate estimators", None, is_complex=True)
    paramSpace = Param("paramSpace", "hyperparameter space", None,
                       is_complex=True)

    def __init__(self, models=None, paramSpace=None, **kwargs):
        super().__init__(**kwargs)
        if models is not None:
            self.set("models", models)
        if paramSpace is not None:
            self._space = paramSpace
        else:
            self._space = None

    def _fit(self, df: pd.DataFrame):
        rng = np.random.default_rng(self.get("seed"))
        models = self.get("models")
        space = self._space or {}
        metric = self.get("evaluationMetric")
        higher_better = metric in ("AUC", "accuracy", "precision", "recall",
                                   "R^2")
        label = self.get("labelCol")
        folds = self.get("numFolds")
        n = len(df)
        fold_id = rng.integers(0, folds, size=n)

        configs = []
        if self.get("searchMode") == "grid":
            import itertools
            keys = [k for k, d in space.items()
                    if isinstance(d, DiscreteHyperParam)]
            grids = [space[k].values for k in keys]
            for base in models:
                for combo in itertools.product(*grids) if grids else [()]:
                    params = {k: v for k, v in zip(keys, combo)
                              if base.hasParam(k)}
                    configs.append((base, params))
        else:
            for _ in range(self.get("numRuns")):
                base = models[int(rng.integers(0, len(models)))]
                params = {k: dist.sample(rng) for k, dist in space.items()
                          if base.hasParam(k)}
                configs.append((base, params))

        def run(cfg):
            base, params = cfg
            scores = []
            for f in range(folds):
                train = df[fold_id != f]
                val = df[fold_id == f]
                est = base.copy(params)
                m = est.fit(train)
                scores.append(_eval_model(m, val, label, metric,
                                          higher_better or metric == "accuracy"))
            return float(np.mean(scores))

        with cf.ThreadPoolExecutor(self.get("parallelism")) as pool:
            scores = list(pool.map(run, configs))
        order = np.argsort(scores)
        best_i = int(order[-1] if higher_better else order[0])
        base, params = configs[best_i]
        best_model = base.copy(params).fit(df)
        out = TuneHyperparametersModel(best=best_model)
        out.set("bestMetric", float(scores[best_i]))
        out.set("bestParams", {k: (v if isinstance(v, (int, float, str, bool))
                                   else str(v)) for k, v in params.items()})
        return out


@register
class TuneHyperparametersModel(Model):
    bestModel = Param("bestModel", "winning fitted model", None, is_complex=True)
    bestMetric = Param("bestMetric", "winning CV metric", None)
    bestParams = Param("bestParams", "winning hyperparameters", None,
                       is_complex=True)

    def __init__(self, best=None, **kwargs):
        super().__init__(**kwargs)
        if best is not None:
            self.set("bestModel", best)

    def _transform(self, df):
        return self.get("bestModel").transform(df)

    def getBestModelInfo(self):
        """Winning hyperparameters + CV metric (TuneHyperparameters.scala
        getBestModelInfo)."""
        return {"params": self.get("bestParams"),
                "metric": self.get("bestMetric")}


@register
class FindBestModel(Estimator):
    """Evaluate fitted models on a dataset, keep the best
    (FindBestModel.scala:50)."""
    evaluationMetric = Param("evaluationMetric", "metric name", "AUC", toString)
    labelCol = Param("labelCol", "label column", "label")
    models = Param("models", "candidate FITTED models", None, is_complex=True)

    def __init__(self, models=None, **kwargs):
        super().__init__(**kwargs)
        if models is not None:
            self.set("models", models)

    def _fit(self, df):
        metric = self.get("evaluationMetric")
        higher_better = metric in ("AUC", "accuracy", "precision", "recall",
                                   "R^2")
        rows = []
        scores = []
        for m in self.get("models"):
            s = _eval_model(m, df, self.get("labelCol"), metric, higher_better)
            scores.append(s)
            rows.append({"model": type(m).__name__, "uid": m.uid, metric: s})
        best_i = int(np.argmax(scores) if higher_better else np.argmin(scores))
        best = self.get("models")[best_i]
        out = BestModel(best=best)
        out.set("allModelMetrics", pd.DataFrame(rows))
        out.set("bestModelMetrics", float(scores[best_i]))
        # BestModel.scoredDataset / rocCurve (FindBestModel.scala:134 params)
        scored = best.transform(df)
        out.set("scoredDataset", scored)
        prob_col = next((c for c in ("probability", "rawPrediction")
                         if c in scored.columns), None)
        y = df[self.get("labelCol")].to_numpy()
        if prob_col is not None and set(np.unique(y)) <= {0, 1, 0.0, 1.0}:
            p1 = np.asarray([np.asarray(v).ravel()[-1]
                             for v in scored[prob_col]])
            order = np.argsort(-p1)
            ys = y[order]
            tp = np.cumsum(ys)
            fp = np.cumsum(1 - ys)
            P, N = max(tp[-1], 1), max(fp[-1], 1)
            out.set("rocCurve", pd.DataFrame(
                {"falsePositiveRate": np.concatenate([[0.0], fp / N]),
                 "truePositiveRate": np.concatenate([[0.0], tp / P])}))
        return out


@register
class BestModel(Model):
    bestModel = Param("bestModel", "winning model", None, is_complex=True)
    allModelMetrics = Param("allModelMetrics", "evaluation table", None,
                            is_complex=True)
    bestModelMetrics = Param("bestModelMetrics", "winning metric", None)
    scoredDataset = Param("scoredDataset", "best model's scores on the "
                          "evaluation dataset", None, is_complex=True)
    rocCurve = Param("rocCurve", "ROC points of the best model (binary)",
                     None, is_complex=True)

    def getScoredDataset(self):
        return self.get("scoredDataset")

    def getRocCurve(self):
        return self.get("rocCurve")

    def __init__(self, best=None, **kwargs):
        super().__init__(**kwargs)
        if best is not None:
            self.set("bestModel", best)

    def _transform(self, df):
        return self.get("bestModel").transform(df)

    def getEvaluationResults(self):
        return self.get("allModelMetrics")
