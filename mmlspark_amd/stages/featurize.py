"""Featurization estimators (core/featurize parity): Featurize (auto-assemble
mixed columns → one vector), CleanMissingData, ValueIndexer, DataConversion,
CountSelector."""
from __future__ import annotations


import numpy as np
import pandas as pd

from ..core.param import Param, toBool, toInt, toList, toString
from ..core.pipeline import Estimator, Model, Transformer
from ..core.registry import register
from ..core.schema import matrix_to_vector_column
from ..models.vw.murmur import hash_string


@register
class Featurize(Estimator):
    """Auto-assemble mixed-type columns into one numeric vector
    (core/.../featurize/Featurize.scala:36): numerics pass through, strings
    one-hot (low cardinality) or hash, missing imputed."""
    inputCols = Param("inputCols", "columns to featurize", None, toList)
    outputCol = Param("outputCol", "feature vector column", "features")
    oneHotEncodeCategoricals = Param("oneHotEncodeCategoricals", "one-hot "
                                     "low-cardinality strings", True, toBool)
    numFeatures = Param("numFeatures", "hash dim for high-cardinality", 262144,
                        toInt)

    def _fit(self, df: pd.DataFrame):
        plan = []
        for c in self.get("inputCols"):
            s = df[c]
            if pd.api.types.is_numeric_dtype(s):
                plan.append({"col": c, "kind": "num",
                             "fill": float(s.mean()) if len(s) else 0.0})
            else:
                cats = [str(x) for x in pd.unique(s.dropna())]
                if self.get("oneHotEncodeCategoricals") and len(cats) <= 100:
                    plan.append({"col": c, "kind": "onehot", "cats": cats})
                else:
                    plan.append({"col": c, "kind": "hash",
                                 "dim": min(self.get("numFeatures"), 4096)})
        m = FeaturizeModel()
        m.set("plan", plan)
        m.set("outputCol", self.get("outputCol"))
        return m


@register
class FeaturizeModel(Model):
    plan = Param("plan", "per-column featurization plan", None, is_complex=True)
    outputCol = Param("outputCol", "feature vector column", "features")

    def _transform(self, df: pd.DataFrame) -> pd.DataFrame:
        parts = []
        for p in self.get("plan"):
            s = df[p["col"]]
            if p["kind"] == "num":
                parts.append(s.fillna(p["fill"]).to_numpy(np.float32)[:, None])
            elif p["kind"] == "onehot":
                cats = {c: i for i, c in enumerate(p["cats"])}
                mat = np.zeros((len(df), len(cats)), dtype=np.float32)
                for r, v in enumerate(s):
                    i = cats.get(str(v))
                    if i is not None:
                        mat[r, i] = 1.0
                parts.append(mat)
            else:
                dim = p["dim"]
                mat = np.zeros((len(df), dim), dtype=np.float32)
                for r, v in enumerate(s):
                    if v is not None and not (isinstance(v, float) and np.isnan(v)):
                        mat[r, hash_string(str(v)) % dim] += 1.0
                parts.append(mat)
        full = np.concatenate(parts, axis=1) if parts else np.zeros((len(df), 0),
                                                                    np.float32)
        out = df.copy()
        out[self.get("outputCol")] = matrix_to_vector_column(full)
        return out


@register
class CleanMissingData(Estimator):
    """Impute missing values: Mean/Median/Custom (CleanMissingData.scala)."""
    inputCols = Param("inputCols", "columns to clean", None, toList)
    outputCols = Param("outputCols", "output columns", None, toList)
    cleaningMode = Param("cleaningMode", "Mean|Median|Custom", "Mean", toString)
    customValue = Param("customValue", "fill value in Custom mode", None)

    def _fit(self, df):
        mode = self.get("cleaningMode")
        fills = {}
        for c in self.get("inputCols"):
            if mode == "Mean":
                fills[c] = float(df[c].mean())
            elif mode == "Median":
                fills[c] = float(df[c].median())
            else:
                fills[c] = float(self.get("customValue"))
        m = CleanMissingDataModel()
        m.set("fillValues", fills)
        m.set("inputCols", self.get("inputCols"))
        m.set("outputCols", self.get("outputCols") or self.get("inputCols"))
        return m


@register
class CleanMissingDataModel(Model):
    inputCols = Param("inputCols", "columns to clean", None, toList)
    outputCols = Param("outputCols", "output columns", None, toList)
    fillValues = Param("fillValues", "per-column fill", None, is_complex=True)

    def _transform(self, df):
        out = df.copy()
        fills = self.get("fillValues")
        for ic, oc in zip(self.get("inputCols"), self.get("outputCols")):
            out[oc] = df[ic].fillna(fills[ic])
        return out


@register
class ValueIndexer(Estimator):
    """Categorical value → index with metadata for inverse (ValueIndexer.scala)."""
    inputCol = Param("inputCol", "column to index", None)
    outputCol = Param("outputCol", "indexed column", None)

    def _fit(self, df):
        vals = sorted((str(v) for v in pd.unique(df[self.get("inputCol")].dropna())))
        m = ValueIndexerModel()
        m.set("levels", list(vals))
        m.set("inputCol", self.get("inputCol"))
        m.set("outputCol", self.get("outputCol") or self.get("inputCol") + "_idx")
        return m


@register
class ValueIndexerModel(Model):
    inputCol = Param("inputCol", "column to index", None)
    outputCol = Param("outputCol", "indexed column", None)
    levels = Param("levels", "ordered category levels", None, toList)

    def _transform(self, df):
        lut = {v: i for i, v in enumerate(self.get("levels"))}
        out = df.copy()
        out[self.get("outputCol")] = [lut.get(str(v), -1)
                                      for v in df[self.get("inputCol")]]
        return out

    def inverse(self, df, col, out_col):
        levels = self.get("levels")
        out = df.copy()
        out[out_col] = [levels[int(i)] if 0 <= int(i) < len(levels) else None
                        for i in df[col]]
        return out


@register
class IndexToValue(Transformer):
    """Inverse of ValueIndexer on a previously indexed column."""
    inputCol = Param("inputCol", "indexed column", None)
    outputCol = Param("outputCol", "value column", None)
    levels = Param("levels", "ordered category levels", None, toList)

    def _transform(self, df):
        levels = self.get("levels") or []
        out = df.copy()
        out[self.get("outputCol")] = [
            levels[int(i)] if 0 <= int(i) < len(levels) else None
            for i in df[self.get("inputCol")]]
        return out


@register
class DataConversion(Transformer):
    """Column dtype conversion (DataConversion.scala)."""
    cols = Param("cols", "columns to convert", None, toList)
    convertTo = Param("convertTo", "boolean|byte|short|integer|long|float|"
                      "double|string|date", "double", toString)

    _MAP = {"boolean": bool, "byte": np.int8, "short": np.int16,
            "integer": np.int32, "long": np.int64, "float": np.float32,
            "double": np.float64, "string": str}

    def _transform(self, df):
        out = df.copy()
        t = self.get("convertTo")
        for c in self.get("cols"):
            if t == "date":
                out[c] = pd.to_datetime(df[c])
            else:
                out[c] = df[c].astype(self._MAP[t])
        return out


@register
class CountSelector(Estimator):
    """Drop all-zero / constant-zero vector slots (CountSelector.scala)."""
    inputCol = Param("inputCol", "vector column", "features")
    outputCol = Param("outputCol", "output column", "features")

    def _fit(self, df):
        mat = np.stack([np.asarray(v, dtype=np.float64)
                        for v in df[self.get("inputCol")]])
        keep = np.nonzero((mat != 0).any(axis=0))[0]
        m = CountSelectorModel()
        m.set("indices", keep.astype(np.int64))
        m.set("inputCol", self.get("inputCol"))
        m.set("outputCol", self.get("outputCol"))
        return m


@register
class CountSelectorModel(Model):
    inputCol = Param("inputCol", "vector column", "features")
    outputCol = Param("outputCol", "output column", "features")
    indices = Param("indices", "kept slots", None, is_complex=True)

    def _transform(self, df):
        keep = np.asarray(self.get("indices"))
        out = df.copy()
        out[self.get("outputCol")] = [np.asarray(v, dtype=np.float32)[keep]
                                      for v in df[self.get("inputCol")]]
        return out


@register
class FastVectorAssembler(Transformer):
    """Assemble scalar + vector columns into one vector column without a
    per-row metadata scan (org/apache/spark/ml/feature/FastVectorAssembler.scala,
    151 LoC — the reference's VectorAssembler fork that skips attribute-group
    rebuilding).  Here: numeric columns copy straight into a preallocated
    float32 matrix; dense-vector columns are stacked; SparseVector columns
    scatter into their slice.  NaN handling follows the reference: rows keep
    their NaNs (no drop)."""
    inputCols = Param("inputCols", "columns to assemble", None, toList)
    outputCol = Param("outputCol", "assembled vector column", "features")

    def _transform(self, df: pd.DataFrame) -> pd.DataFrame:
        from ..core.schema import SparseVector
        cols = self.get("inputCols")
        n = len(df)
        widths = []
        for c in cols:
            s = df[c]
            if pd.api.types.is_numeric_dtype(s):
                widths.append(1)
            else:
                v0 = s.dropna().iloc[0] if s.notna().any() else None
                widths.append(v0.size if isinstance(v0, SparseVector)
                              else (len(np.asarray(v0)) if v0 is not None else 0))
        total = int(sum(widths))
        mat = np.zeros((n, total), dtype=np.float32)
        off = 0
        for c, w in zip(cols, widths):
            s = df[c]
            if pd.api.types.is_numeric_dtype(s):
                mat[:, off] = s.to_numpy(np.float32)
            else:
                vals = s.to_numpy()
                first = next((v for v in vals if v is not None), None)
                if isinstance(first, SparseVector):
                    for r, v in enumerate(vals):
                        if v is not None:
                            mat[r, off + v.indices] = v.values
                elif first is not None:
                    mat[:, off:off + w] = np.stack(
                        [np.full(w, np.nan, np.float32) if v is None
                         else np.asarray(v, dtype=np.float32) for v in vals])
            off += w
        out = df.copy()
        out[self.get("outputCol")] = matrix_to_vector_column(mat)
        return out
