"""DataFrame/vector-column helpers.

The framework's interchange format is a pandas DataFrame; a "vector column"
is an object column of equal-length 1-D float arrays (analog of SparkML's
VectorUDT columns consumed everywhere in the reference). These helpers
convert between vector columns and dense 2-D numpy/torch matrices, and pick
unused column names (analog of DatasetExtensions.findUnusedColumnName,
core/.../core/schema/DatasetExtensions.scala).
"""
from __future__ import annotations

from typing import List, Optional, Sequence

import numpy as np
import pandas as pd


class SparseVector:
    """Hashed sparse feature vector (the analog of SparkML SparseVector
    produced by the reference's VowpalWabbitFeaturizer)."""

    __slots__ = ("size", "indices", "values")

    def __init__(self, size: int, indices, values):
        self.size = int(size)
        self.indices = np.asarray(indices, dtype=np.int32)
        self.values = np.asarray(values, dtype=np.float32)

    def to_dense(self) -> np.ndarray:
        out = np.zeros(self.size, dtype=np.float32)
        out[self.indices] = self.values
        return out

    def __len__(self):
        return self.size

    def __eq__(self, other):
        return (isinstance(other, SparseVector) and self.size == other.size
                and np.array_equal(self.indices, other.indices)
                and np.array_equal(self.values, other.values))

    def __repr__(self):
        return f"SparseVector(size={self.size}, nnz={len(self.indices)})"


def find_unused_column(df: pd.DataFrame, base: str) -> str:
    name = base
    i = 0
    while name in df.columns:
        i += 1
        name = f"{base}_{i}"
    return name


def is_vector_column(df: pd.DataFrame, col: str) -> bool:
    if col not in df.columns or len(df) == 0:
        return False
    v = df[col].iloc[0]
    return isinstance(v, (np.ndarray, list, tuple))


def vector_column_to_matrix(df: pd.DataFrame, col: str,
                            dtype=np.float32) -> np.ndarray:
    """Object column of arrays -> dense (n, d) matrix."""
    vals = df[col].to_numpy()
    if len(vals) == 0:
        return np.zeros((0, 0), dtype=dtype)
    first = np.asarray(vals[0], dtype=dtype)
    out = np.empty((len(vals), first.shape[0]), dtype=dtype)
    for i, v in enumerate(vals):
        out[i] = np.asarray(v, dtype=dtype)
    return out


def matrix_to_vector_column(mat: np.ndarray) -> list:
    """Dense (n, d) matrix -> list of 1-D arrays for an object column.
    list() iterates at C level and row views need no copy."""
    return list(np.asarray(mat))


def features_matrix(df: pd.DataFrame, features_col: str = "features",
                    feature_cols: Optional[Sequence[str]] = None,
                    dtype=np.float32) -> np.ndarray:
    """Resolve features: either a single vector column, or a list of numeric cols."""
    if feature_cols:
        return df[list(feature_cols)].to_numpy(dtype=dtype)
    if is_vector_column(df, features_col):
        return vector_column_to_matrix(df, features_col, dtype=dtype)
    if features_col in df.columns:
        return df[[features_col]].to_numpy(dtype=dtype)
    raise KeyError(
        f"no features: column {features_col!r} absent and no featureCols given "
        f"(columns: {list(df.columns)})")


def infer_feature_names(df: pd.DataFrame, features_col: str,
                        feature_cols: Optional[Sequence[str]]) -> List[str]:
    if feature_cols:
        return list(feature_cols)
    if len(df) and features_col in df.columns \
            and isinstance(df[features_col].iloc[0], SparseVector):
        d = df[features_col].iloc[0].size
        return [f"{features_col}_{i}" for i in range(d)]
    if is_vector_column(df, features_col):
        d = len(np.asarray(df[features_col].iloc[0])) if len(df) else 0
        return [f"{features_col}_{i}" for i in range(d)]
    return [features_col]
