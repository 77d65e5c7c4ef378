"""Arrow / Spark DataFrame interop at the estimator boundary (VERDICT r1
item 2; the reference's surface is Spark DataFrames —
LightGBMBase.scala:480-484, IOImplicits.scala:22-59).

pyarrow tests always run; pyspark tests run when pyspark is installed
(skipped in this image — no pyspark wheel offline) but the code path is
identical: kind detection → pandas coercion → restore on output.
"""
import numpy as np
import pandas as pd
import pytest

from mmlspark_amd.core.interop import (arrow_to_pandas, coerce_input,
                                       pandas_to_arrow, restore_output)
from mmlspark_amd.core.schema import SparseVector
from mmlspark_amd.models.gbdt.estimators import LightGBMClassifier

pa = pytest.importorskip("pyarrow")


def _df(n=600, nf=8, seed=0):
    rng = np.random.default_rng(seed)
    X = rng.normal(size=(n, nf)).astype(np.float32)
    y = (X[:, 0] + X[:, 1] > 0).astype(np.float64)
    return pd.DataFrame({"features": list(X), "label": y})


def test_arrow_round_trip_vector_columns():
    df = _df()
    t = pandas_to_arrow(df)
    assert isinstance(t, pa.Table)
    assert pa.types.is_list(t.schema.field("features").type)
    back = arrow_to_pandas(t)
    np.testing.assert_allclose(np.stack(back["features"].to_numpy()),
                               np.stack(df["features"].to_numpy()))
    np.testing.assert_allclose(back["label"].to_numpy(),
                               df["label"].to_numpy())


def test_arrow_round_trip_sparse_vectors():
    rows = [SparseVector(10, [1, 4], [1.0, -2.0]),
            SparseVector(10, [0], [3.0])]
    df = pd.DataFrame({"features": rows, "label": [1.0, 0.0]})
    back = arrow_to_pandas(pandas_to_arrow(df))
    v = back["features"].iloc[0]
    assert isinstance(v, SparseVector)
    assert v.size == 10 and v.indices.tolist() == [1, 4]


def test_fit_transform_accepts_arrow_table():
    """Arrow in → Arrow out, numerically identical to the pandas path."""
    df = _df()
    table = pandas_to_arrow(df)
    m_arrow = LightGBMClassifier(numIterations=8, numLeaves=15, seed=3).fit(table)
    m_pd = LightGBMClassifier(numIterations=8, numLeaves=15, seed=3).fit(df)
    assert m_arrow.booster.save_to_string() == m_pd.booster.save_to_string()

    out = m_arrow.transform(table)
    assert isinstance(out, pa.Table)
    out_pd = m_pd.transform(df)
    np.testing.assert_allclose(
        np.stack(arrow_to_pandas(out)["probability"].to_numpy()),
        np.stack(out_pd["probability"].to_numpy()), atol=1e-6)


def test_fit_transform_save_load_arrow(tmp_path):
    df = _df(seed=5)
    table = pandas_to_arrow(df)
    m = LightGBMClassifier(numIterations=5, numLeaves=7).fit(table)
    p = str(tmp_path / "m")
    m.save(p)
    from mmlspark_amd.core.serialize import load_stage
    m2 = load_stage(p)
    out1 = arrow_to_pandas(m.transform(table))
    out2 = arrow_to_pandas(m2.transform(table))
    np.testing.assert_allclose(np.stack(out1["probability"].to_numpy()),
                               np.stack(out2["probability"].to_numpy()))


def test_coerce_kind_detection():
    df = _df(n=10)
    _, k1 = coerce_input(df)
    assert k1 == "pandas"
    _, k2 = coerce_input(pandas_to_arrow(df))
    assert k2 == "arrow"
    assert restore_output(df, "pandas") is df


def test_pipeline_with_arrow_input():
    from mmlspark_amd.core.pipeline import Pipeline
    from mmlspark_amd.stages.basic import DropColumns
    df = _df(n=100)
    df["junk"] = 1.0
    table = pandas_to_arrow(df)
    pipe = Pipeline([DropColumns(cols=["junk"]),
                     LightGBMClassifier(numIterations=3, numLeaves=7)])
    pm = pipe.fit(table)
    out = pm.transform(table)
    assert isinstance(out, pa.Table)
    assert "prediction" in out.column_names


spark = None
try:  # pragma: no cover - pyspark absent in this image
    import pyspark  # noqa: F401
    spark = True
except ImportError:
    pass


@pytest.mark.skipif(spark is None, reason="pyspark not installed")
def test_fit_transform_accepts_spark_dataframe():  # pragma: no cover
    """BASELINE config #1 literal: Spark local[2], CPU plumbing."""
    from pyspark.sql import SparkSession
    ss = (SparkSession.builder.master("local[2]")
          .appName("mmlspark_amd-interop").getOrCreate())
    df = _df()
    from mmlspark_amd.core.interop import pandas_to_spark
    sdf = pandas_to_spark(df, ss)
    m_spark = LightGBMClassifier(numIterations=8, numLeaves=15, seed=3).fit(sdf)
    m_pd = LightGBMClassifier(numIterations=8, numLeaves=15, seed=3).fit(df)
    assert m_spark.booster.save_to_string() == m_pd.booster.save_to_string()
    out = m_spark.transform(sdf)
    assert out.__class__.__module__.startswith("pyspark.sql")
    assert "probability" in out.columns


def test_arrow_nulls_and_chunked_tables():
    """Null list entries fall back to the slow path (None preserved);
    multi-chunk tables combine correctly through the flat-buffer path."""
    rows = [np.arange(4, dtype=np.float32), None,
            np.ones(4, dtype=np.float32)]
    t = pa.table({"features": pa.array(
        [None if r is None else r.tolist() for r in rows],
        type=pa.list_(pa.float32()))})
    back = arrow_to_pandas(t)
    assert back["features"].iloc[1] is None
    np.testing.assert_allclose(back["features"].iloc[0], rows[0])

    # chunked: two record batches concatenated
    df = pd.DataFrame({"features": [np.full(3, i, dtype=np.float32)
                                    for i in range(10)]})
    t1 = pandas_to_arrow(df.iloc[:6])
    t2 = pandas_to_arrow(df.iloc[6:])
    both = pa.concat_tables([t1, t2])
    assert both.column(0).num_chunks == 2
    back = arrow_to_pandas(both)
    assert len(back) == 10
    np.testing.assert_allclose(back["features"].iloc[9], np.full(3, 9.0))


def test_arrow_image_schema_struct_decodes():
    """Spark ImageSchema structs (origin/height/width/nChannels/mode/data)
    arriving through Arrow become HWC uint8 arrays the image stages use
    (ImageSchemaUtils parity)."""
    rng = np.random.default_rng(0)
    imgs = [rng.integers(0, 255, (8, 6, 3)).astype(np.uint8)
            for _ in range(3)]
    rows = [{"origin": f"mem://{i}", "height": 8, "width": 6,
             "nChannels": 3, "mode": 16,
             "data": im.tobytes()} for i, im in enumerate(imgs)]
    t = pa.table({"image": pa.array(rows)})
    back = arrow_to_pandas(t)
    got = back["image"].iloc[1]
    assert got.shape == (8, 6, 3) and got.dtype == np.uint8
    np.testing.assert_array_equal(got, imgs[1])
    # flows straight into ImageTransformer
    from mmlspark_amd.models.images import ImageTransformer
    out = ImageTransformer(inputCol="image", outputCol="resized") \
        .resize(4, 4).transform(back)
    assert out["resized"].iloc[0].shape[:2] == (4, 4)


def test_arrow_image_round_trip():
    """uint8 HWC image columns survive pandas → Arrow → pandas unchanged
    (encoded as ImageSchema structs on the way out)."""
    rng = np.random.default_rng(1)
    imgs = [rng.integers(0, 255, (5, 7, 3)).astype(np.uint8)
            for _ in range(4)]
    df = pd.DataFrame({"image": imgs, "k": list(range(4))})
    t = pandas_to_arrow(df)
    assert pa.types.is_struct(t.schema.field("image").type)
    back = arrow_to_pandas(t)
    for a, b in zip(back["image"], imgs):
        np.testing.assert_array_equal(a, b)
