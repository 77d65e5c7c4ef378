"""Generic DataFrame transformer stages (core/.../stages/ parity):
DropColumns, SelectColumns, RenameColumn, Repartition, UDFTransformer,
Lambda, EnsembleByKey, Explode, Cacher, Timer, MultiColumnAdapter,
SummarizeData, TextPreprocessor, UnicodeNormalize."""
from __future__ import annotations

import time
import unicodedata
from typing import Callable, Optional

import numpy as np
import pandas as pd

from ..core.param import Param, toBool, toInt, toList, toString
from ..core.pipeline import Estimator, Model, Transformer
from ..core.registry import register


@register
class DropColumns(Transformer):
    cols = Param("cols", "columns to drop", None, toList)

    def _transform(self, df):
        return df.drop(columns=[c for c in (self.get("cols") or [])
                                if c in df.columns])


@register
class SelectColumns(Transformer):
    cols = Param("cols", "columns to keep", None, toList)

    def _transform(self, df):
        return df[self.get("cols") or []].copy()


@register
class RenameColumn(Transformer):
    inputCol = Param("inputCol", "column to rename", None)
    outputCol = Param("outputCol", "new name", None)

    def _transform(self, df):
        return df.rename(columns={self.get("inputCol"): self.get("outputCol")})


@register
class Repartition(Transformer):
    """Partition-count hint; here partitions are GPU-rank shards, so this is
    a shuffling no-op that records the requested parallelism (Repartition.scala)."""
    n = Param("n", "number of partitions", 1, toInt)
    disable = Param("disable", "pass-through", False, toBool)

    def _transform(self, df):
        return df


@register
class UDFTransformer(Transformer):
    inputCol = Param("inputCol", "input column", None)
    inputCols = Param("inputCols", "input columns", None)
    outputCol = Param("outputCol", "output column", "output")

    def __init__(self, udf: Optional[Callable] = None, **kwargs):
        super().__init__(**kwargs)
        self._udf = udf

    def setUDF(self, fn):
        self._udf = fn
        return self

    def _transform(self, df):
        out = df.copy()
        if self.get("inputCols"):
            cols = self.get("inputCols")
            out[self.get("outputCol")] = [
                self._udf(*vals) for vals in zip(*[df[c] for c in cols])]
        else:
            out[self.get("outputCol")] = df[self.get("inputCol")].map(self._udf)
        return out


@register
class Lambda(Transformer):
    """Arbitrary DataFrame → DataFrame function (Lambda.scala:22)."""

    def __init__(self, fn: Optional[Callable] = None, **kwargs):
        super().__init__(**kwargs)
        self._fn = fn

    def setTransform(self, fn):
        self._fn = fn
        return self

    def _transform(self, df):
        return self._fn(df)


@register
class Explode(Transformer):
    inputCol = Param("inputCol", "list column to explode", None)
    outputCol = Param("outputCol", "exploded column", "output")

    def _transform(self, df):
        out = df.copy()
        out[self.get("outputCol")] = df[self.get("inputCol")]
        return out.explode(self.get("outputCol")).reset_index(drop=True)


@register
class Cacher(Transformer):
    """Materialization hint (Cacher.scala) — pandas frames are eager; copy."""
    disable = Param("disable", "pass-through", False, toBool)

    def _transform(self, df):
        return df if self.get("disable") else df.copy()


@register
class Timer(Transformer):
    """Times an inner stage (Timer.scala); logs to the telemetry ring."""
    stage = Param("stage", "inner stage", None, is_complex=True)
    logToScala = Param("logToScala", "log to driver", True, toBool)
    disableMaterialization = Param("disableMaterialization", "skip forcing "
                                   "the inner result (no-op: pandas frames "
                                   "are eager — Timer.scala parity)", True,
                                   toBool)

    def _transform(self, df):
        from ..core.telemetry import log_stage_event
        inner = self.get("stage")
        t0 = time.perf_counter()
        out = inner.transform(df)
        log_stage_event(self, "timer", inner=type(inner).__name__,
                        ms=(time.perf_counter() - t0) * 1e3)
        return out


@register
class EnsembleByKey(Transformer):
    """Average vector/scalar columns grouped by key columns (EnsembleByKey.scala)."""
    keys = Param("keys", "group-by key columns", None, toList)
    cols = Param("cols", "columns to average", None, toList)
    strategy = Param("strategy", "mean", "mean", toString)
    collapseGroup = Param("collapseGroup", "one row per group", True, toBool)

    def _transform(self, df):
        keys = self.get("keys")
        cols = self.get("cols")

        def agg(series):
            vals = series.to_numpy()
            if len(vals) and isinstance(vals[0], (np.ndarray, list)):
                return np.mean([np.asarray(v, dtype=np.float64) for v in vals],
                               axis=0)
            return float(np.mean(vals))

        grouped = df.groupby(keys, sort=False)
        rows = []
        for key, g in grouped:
            row = dict(zip(keys, key if isinstance(key, tuple) else (key,)))
            for c in cols:
                row[f"{c}_ensemble" if not c.endswith("_ensemble") else c] = agg(g[c])
            rows.append(row)
        res = pd.DataFrame(rows)
        if not self.get("collapseGroup"):
            res = df.merge(res, on=keys, how="left")
        return res


@register
class MultiColumnAdapter(Transformer):
    """Apply a single-column stage to many columns (MultiColumnAdapter.scala)."""
    baseStage = Param("baseStage", "stage with inputCol/outputCol", None,
                      is_complex=True)
    inputCols = Param("inputCols", "input columns", None, toList)
    outputCols = Param("outputCols", "output columns", None, toList)

    def _transform(self, df):
        out = df
        for ic, oc in zip(self.get("inputCols"), self.get("outputCols")):
            stage = self.get("baseStage").copy()
            stage.uid = stage.uid + "_" + ic
            stage.set("inputCol", ic)
            stage.set("outputCol", oc)
            out = stage.transform(out)
        return out


@register
class SummarizeData(Transformer):
    """Column statistics summary (SummarizeData.scala: counts/quantiles/basic)."""
    counts = Param("counts", "include counts", True, toBool)
    basic = Param("basic", "include basic stats", True, toBool)
    percentiles = Param("percentiles", "include percentiles", True, toBool)

    def _transform(self, df):
        rows = []
        for c in df.columns:
            s = df[c]
            row = {"Feature": c}
            if self.get("counts"):
                row.update({"Count": float(len(s)),
                            "Unique Value Count": float(s.nunique()),
                            "Missing Value Count": float(s.isna().sum())})
            if pd.api.types.is_numeric_dtype(s):
                if self.get("basic"):
                    row.update({"Mean": float(s.mean()), "Std": float(s.std()),
                                "Min": float(s.min()), "Max": float(s.max())})
                if self.get("percentiles"):
                    for q in (0.005, 0.01, 0.05, 0.25, 0.5, 0.75, 0.95, 0.99,
                              0.995):
                        row[f"P{q}"] = float(s.quantile(q))
            rows.append(row)
        return pd.DataFrame(rows)


@register
class TextPreprocessor(Transformer):
    """Map/normalize text via a substitution dict then trie-based replace
    (TextPreprocessor.scala)."""
    inputCol = Param("inputCol", "text column", "text")
    outputCol = Param("outputCol", "output column", "output")
    normFunc = Param("normFunc", "lowercase|identity", "lowercase", toString)
    map = Param("map", "substring substitution map", None)

    def _transform(self, df):
        sub = self.get("map") or {}
        norm = (lambda s: s.lower()) if self.get("normFunc") == "lowercase" \
            else (lambda s: s)

        def proc(s):
            s = norm(str(s))
            for k in sorted(sub, key=len, reverse=True):
                s = s.replace(norm(k), sub[k])
            return s
        out = df.copy()
        out[self.get("outputCol")] = df[self.get("inputCol")].map(proc)
        return out


@register
class UnicodeNormalize(Transformer):
    inputCol = Param("inputCol", "text column", "text")
    outputCol = Param("outputCol", "output column", "output")
    form = Param("form", "NFC|NFD|NFKC|NFKD", "NFKD", toString)
    lower = Param("lower", "lowercase", True, toBool)

    def _transform(self, df):
        def proc(s):
            s = unicodedata.normalize(self.get("form"), str(s))
            return s.lower() if self.get("lower") else s
        out = df.copy()
        out[self.get("outputCol")] = df[self.get("inputCol")].map(proc)
        return out


@register
class ClassBalancer(Estimator):
    """Per-class balancing weights: weight = max(count)/count(class)
    (stages/ClassBalancer.scala)."""
    inputCol = Param("inputCol", "label column", "label")
    outputCol = Param("outputCol", "weight column", "weight")
    broadcastJoin = Param("broadcastJoin", "join strategy hint", True, toBool)

    def _fit(self, df):
        counts = df[self.get("inputCol")].value_counts()
        mx = counts.max()
        m = ClassBalancerModel()
        m.set("weights", {str(k): float(mx / v) for k, v in counts.items()})
        m.set("inputCol", self.get("inputCol"))
        m.set("outputCol", self.get("outputCol"))
        return m


@register
class ClassBalancerModel(Model):
    inputCol = Param("inputCol", "label column", "label")
    outputCol = Param("outputCol", "weight column", "weight")
    weights = Param("weights", "class → weight", None, is_complex=True)

    def _transform(self, df):
        w = self.get("weights")
        out = df.copy()
        out[self.get("outputCol")] = [w.get(str(v), 1.0)
                                      for v in df[self.get("inputCol")]]
        return out


@register
class StratifiedRepartition(Transformer):
    """Reorder rows so every equal-size shard (GPU rank) sees the same label
    mix (stages/StratifiedRepartition.scala: class-balanced partitions)."""
    labelCol = Param("labelCol", "label column", "label")
    mode = Param("mode", "equal|original|mixed", "equal", toString)
    seed = Param("seed", "shuffle seed", 0, toInt)

    def _transform(self, df):
        rng = np.random.default_rng(self.get("seed"))
        groups = [g.sample(frac=1.0, random_state=int(rng.integers(1 << 31)))
                  for _, g in df.groupby(self.get("labelCol"), sort=False)]
        # round-robin interleave classes so contiguous shards are stratified
        idx_lists = [list(g.index) for g in groups]
        order = []
        while any(idx_lists):
            for lst in idx_lists:
                if lst:
                    order.append(lst.pop(0))
        return df.loc[order].reset_index(drop=True)
