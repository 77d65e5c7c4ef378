"""LocalExplainer scaffold (core/.../explainers/LocalExplainer.scala:16).

The perturbation fan-out batches ALL samples for a chunk of rows into one
DataFrame and scores it with a single model.transform — on GPU this drives
the batched HIP scoring kernels (the reference instead exploded rows into
Spark and scored per-row UDFs)."""
from __future__ import annotations


import numpy as np
import pandas as pd

from ..core.param import Param, toInt, toList
from ..core.pipeline import Transformer


class LocalExplainer(Transformer):
    model = Param("model", "the model Transformer to explain", None,
                  is_complex=True)
    targetCol = Param("targetCol", "model output column to explain",
                      "probability")
    targetClasses = Param("targetClasses", "class indices to explain", [1],
                          toList)
    outputCol = Param("outputCol", "explanation output column", "explanation")
    numSamples = Param("numSamples", "perturbations per row (default: "
                       "modality-specific)", None)
    metricsCol = Param("metricsCol", "fit-metric output column (r2)", "r2")
    seed = Param("seed", "sampling seed", 0, toInt)
    rowBatch = Param("rowBatch", "rows explained per scoring fan-out", 16, toInt)

    def _score_samples(self, samples_df: pd.DataFrame) -> np.ndarray:
        """Run the inner model; return (n, n_target_classes) target values."""
        model = self.get("model")
        out = model.transform(samples_df)
        col = out[self.get("targetCol")].to_numpy()
        classes = self.get("targetClasses")
        if len(col) and isinstance(col[0], (np.ndarray, list)):
            mat = np.stack([np.asarray(v) for v in col])
            return mat[:, classes]
        return np.asarray(col, dtype=np.float64)[:, None]

    def _score_matrix(self, X: np.ndarray, cols=None) -> np.ndarray:
        """Score a raw feature MATRIX.  Models exposing score_matrix (the
        GBDT models) skip the whole DataFrame round trip — the fan-out
        builds millions of object cells otherwise; generic transformers
        fall back to _score_samples."""
        model = self.get("model")
        if hasattr(model, "score_matrix"):
            v = model.score_matrix(X, self.get("targetCol"),
                                   self.get("targetClasses"))
            if v is not None:
                return v
        if cols is not None:
            return self._score_samples(pd.DataFrame(X, columns=list(cols)))
        from ..core.schema import matrix_to_vector_column
        fcol = model.get("featuresCol") if "featuresCol" in model.params() \
            else "features"
        return self._score_samples(pd.DataFrame({
            fcol: matrix_to_vector_column(X.astype(np.float32))}))
