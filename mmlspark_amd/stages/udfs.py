"""Column helper functions (stages/udfs.scala parity): to_vector,
get_value_at, vector_to_array — plus the FluentAPI chaining helpers
(core/src/main/python/mmlspark/core/spark/FluentAPI.py parity)."""
from __future__ import annotations

import numpy as np
import pandas as pd


def to_vector(df: pd.DataFrame, cols, output_col: str = "features",
              dtype=np.float32) -> pd.DataFrame:
    """Assemble numeric columns into a dense vector column."""
    out = df.copy()
    mat = df[list(cols)].to_numpy(dtype=dtype)
    out[output_col] = [row for row in mat]
    return out


def get_value_at(df: pd.DataFrame, vector_col: str, index: int,
                 output_col: str) -> pd.DataFrame:
    """Extract element `index` of a vector column."""
    out = df.copy()
    out[output_col] = [float(np.asarray(v)[index]) for v in df[vector_col]]
    return out


def vector_to_array(df: pd.DataFrame, vector_col: str,
                    output_col: str) -> pd.DataFrame:
    out = df.copy()
    out[output_col] = [np.asarray(v, dtype=np.float64).tolist()
                       for v in df[vector_col]]
    return out


def ml_transform(df: pd.DataFrame, *stages) -> pd.DataFrame:
    """Fluent chaining: df |> stage1 |> stage2 (FluentAPI mlTransform)."""
    cur = df
    for st in stages:
        cur = st.transform(cur)
    return cur


def ml_fit(df: pd.DataFrame, estimator):
    """Fluent fit (FluentAPI mlFit)."""
    return estimator.fit(df)


def install_fluent_api():
    """Opt-in monkey patch (core/spark/FluentAPI.py parity): gives every
    pandas DataFrame `.mlTransform(*stages)` and `.mlFit(estimator)` so
    reference-style fluent chains port verbatim."""
    def _ml_transform(self, *stages):
        return ml_transform(self, *stages)

    def _ml_fit(self, estimator):
        return ml_fit(self, estimator)

    pd.DataFrame.mlTransform = _ml_transform
    pd.DataFrame.mlFit = _ml_fit
