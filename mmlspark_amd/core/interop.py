"""DataFrame interop: pyarrow Tables and pyspark DataFrames at the
Estimator/Transformer boundary.

The reference's entire deployment surface is Spark DataFrames
(LightGBMBase.scala:480-484 barrier mapPartitions; IOImplicits.scala:22-59).
Here the native interchange format is pandas (vector columns = object
columns of 1-D float arrays); this module lets every stage accept

  * a pandas DataFrame            (native — passthrough),
  * a pyarrow Table               (list/fixed-size-list columns → vector
                                   columns; SparkML VectorUDT structs →
                                   ndarray / SparseVector),
  * a pyspark DataFrame           (via toPandas + ml.linalg coercion),

and `transform` returns the same kind it was given (Arrow in → Arrow out,
Spark in → Spark out).  The analog of LightGBMBase.prepareDataframe's
column coercion (LightGBMBase.scala:110-145).
"""
from __future__ import annotations

from typing import Tuple

import numpy as np
import pandas as pd

from .schema import SparseVector

PANDAS, ARROW, SPARK = "pandas", "arrow", "spark"


def _is_arrow(obj) -> bool:
    try:
        import pyarrow as pa
    except ImportError:
        return False
    return isinstance(obj, (pa.Table, pa.RecordBatch))


def _is_spark(obj) -> bool:
    m = type(obj).__module__ or ""
    return m.startswith("pyspark.sql")


_IMAGE_FIELDS = {"origin", "height", "width", "nChannels", "mode", "data"}


def _image_struct_to_obj(v):
    """Spark ImageSchema struct → HWC uint8 ndarray (the layout every
    image stage here consumes; ImageSchemaUtils / ImageUtils parity —
    Spark stores BGR row-major bytes)."""
    if v is None:
        return None
    h, w, c = int(v["height"]), int(v["width"]), int(v["nChannels"])
    buf = v["data"]
    arr = np.frombuffer(bytes(buf), dtype=np.uint8)[: h * w * c]
    a = arr.reshape(h, w, c)
    return a[:, :, 0] if c == 1 else a


def _vector_struct_to_obj(v):
    """SparkML VectorUDT struct (dict after Arrow/pandas conversion)."""
    if v is None:
        return None
    t = v.get("type")
    if t == 1 or (t is None and v.get("values") is not None
                  and v.get("indices") is None):
        return np.asarray(v["values"], dtype=np.float32)
    return SparseVector(int(v["size"]),
                        np.asarray(v["indices"], dtype=np.int32),
                        np.asarray(v["values"], dtype=np.float32))


def arrow_to_pandas(table) -> pd.DataFrame:
    import pyarrow as pa
    if isinstance(table, pa.RecordBatch):
        table = pa.Table.from_batches([table])
    out = {}
    for name, col in zip(table.column_names, table.columns):
        typ = col.type
        if pa.types.is_fixed_size_list(typ) or pa.types.is_list(typ) \
                or pa.types.is_large_list(typ):
            arr = col.combine_chunks()
            if arr.null_count == 0 and pa.types.is_floating(typ.value_type) \
                    or (arr.null_count == 0
                        and pa.types.is_integer(typ.value_type)):
                # zero-copy-ish: one flat buffer + per-row views
                flat = np.asarray(arr.values).astype(np.float32, copy=False)
                if pa.types.is_fixed_size_list(typ):
                    rows = list(flat.reshape(-1, typ.list_size))
                else:
                    offs = np.asarray(arr.offsets)
                    rows = [flat[offs[i]:offs[i + 1]]
                            for i in range(len(arr))]
                out[name] = pd.Series(rows, dtype=object)
                continue
            vals = col.to_pylist()
            out[name] = pd.Series(
                [None if v is None else np.asarray(v, dtype=np.float32)
                 for v in vals], dtype=object)
        elif pa.types.is_struct(typ) and {"size", "indices", "values"} <= {
                f.name for f in typ}:
            out[name] = pd.Series(
                [_vector_struct_to_obj(v) for v in col.to_pylist()],
                dtype=object)
        elif pa.types.is_struct(typ) and _IMAGE_FIELDS <= {
                f.name for f in typ}:
            out[name] = pd.Series(
                [_image_struct_to_obj(v) for v in col.to_pylist()],
                dtype=object)
        else:
            out[name] = col.to_pandas()
    return pd.DataFrame(out)


def pandas_to_arrow(df: pd.DataFrame):
    import pyarrow as pa
    arrays, names = [], []
    for name in df.columns:
        s = df[name]
        if s.dtype == object and len(s) and isinstance(
                s.dropna().iloc[0] if s.notna().any() else None,
                (np.ndarray, list, SparseVector)):
            first = s.dropna().iloc[0]
            if isinstance(first, np.ndarray) and first.ndim >= 2 \
                    and first.dtype == np.uint8:
                # image arrays → ImageSchema structs (round-trip symmetry)
                def _to_img(v):
                    if v is None:
                        return None
                    a = v if v.ndim == 3 else v[:, :, None]
                    return {"origin": "", "height": a.shape[0],
                            "width": a.shape[1], "nChannels": a.shape[2],
                            "mode": 16 if a.shape[2] == 3 else 0,
                            "data": a.tobytes()}
                arrays.append(pa.array([_to_img(v) for v in s]))
                names.append(str(name))
                continue
            if isinstance(first, SparseVector):
                arrays.append(pa.array(
                    [None if v is None else
                     {"type": 0, "size": v.size,
                      "indices": v.indices.tolist(),
                      "values": v.values.tolist()} for v in s]))
            else:
                arrays.append(pa.array(
                    [None if v is None else np.asarray(v, dtype=np.float32)
                     for v in s],
                    type=pa.list_(pa.float32())))
        else:
            arrays.append(pa.Array.from_pandas(s))
        names.append(str(name))
    return pa.Table.from_arrays(arrays, names=names)


def spark_to_pandas(sdf) -> pd.DataFrame:
    pdf = sdf.toPandas()
    try:
        from pyspark.ml.linalg import DenseVector, SparseVector as PSV
    except ImportError:  # pyspark without ml — nothing to coerce
        return pdf
    for c in pdf.columns:
        if pdf[c].dtype == object and len(pdf):
            v0 = pdf[c].dropna()
            v0 = v0.iloc[0] if len(v0) else None
            if hasattr(v0, "asDict") and _IMAGE_FIELDS <= set(
                    v0.asDict().keys()):  # ImageSchema Row
                pdf[c] = pdf[c].map(
                    lambda v: None if v is None
                    else _image_struct_to_obj(v.asDict()))
            elif isinstance(v0, DenseVector):
                pdf[c] = pdf[c].map(
                    lambda v: None if v is None
                    else np.asarray(v.toArray(), dtype=np.float32))
            elif isinstance(v0, PSV):
                pdf[c] = pdf[c].map(
                    lambda v: None if v is None
                    else SparseVector(v.size, v.indices, v.values))
    return pdf


def pandas_to_spark(pdf: pd.DataFrame, spark=None):
    from pyspark.ml.linalg import Vectors
    from pyspark.sql import SparkSession
    spark = spark or SparkSession.getActiveSession() or \
        SparkSession.builder.getOrCreate()
    conv = pdf.copy()
    for c in conv.columns:
        if conv[c].dtype == object and len(conv):
            v0 = conv[c].dropna()
            v0 = v0.iloc[0] if len(v0) else None
            if isinstance(v0, np.ndarray):
                conv[c] = conv[c].map(
                    lambda v: None if v is None
                    else Vectors.dense([float(x) for x in v]))
            elif isinstance(v0, SparseVector):
                conv[c] = conv[c].map(
                    lambda v: None if v is None
                    else Vectors.sparse(v.size, v.indices.tolist(),
                                        v.values.tolist()))
    return spark.createDataFrame(conv)


def coerce_input(df) -> Tuple[pd.DataFrame, str]:
    """Any supported DataFrame kind → (pandas, original kind)."""
    if isinstance(df, pd.DataFrame):
        return df, PANDAS
    if _is_arrow(df):
        return arrow_to_pandas(df), ARROW
    if _is_spark(df):
        return spark_to_pandas(df), SPARK
    return df, PANDAS  # unknown: let the stage raise its own error


def restore_output(out, kind: str):
    """transform() result back to the caller's DataFrame kind.

    Columns that have no Arrow/Spark representation (e.g. rich Python
    objects some transformers emit) fall back to pandas with a warning
    rather than failing the whole transform."""
    if kind == PANDAS or not isinstance(out, pd.DataFrame):
        return out
    try:
        if kind == ARROW:
            return pandas_to_arrow(out)
        if kind == SPARK:
            return pandas_to_spark(out)
    except Exception as e:
        import warnings
        warnings.warn(f"could not convert transform output back to {kind} "
                      f"({e!r}); returning pandas", RuntimeWarning)
    return out
