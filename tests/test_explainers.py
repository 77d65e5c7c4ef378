"""KernelSHAP / LIME explainers: additivity, signal recovery, modalities."""
import numpy as np
import pandas as pd
import pytest

from mmlspark_amd.core.schema import matrix_to_vector_column
from mmlspark_amd.explainers.lime import TabularLIME, TextLIME, VectorLIME
from mmlspark_amd.explainers.shap import TabularSHAP, TextSHAP, VectorSHAP
from mmlspark_amd.models.gbdt.estimators import LightGBMClassifier


@pytest.fixture(scope="module")
def model_and_data():
    rng = np.random.default_rng(0)
    n, nf = 2000, 5
    X = rng.normal(size=(n, nf)).astype(np.float32)
    # only features 0 and 1 matter
    y = ((2 * X[:, 0] - 1.5 * X[:, 1] + rng.normal(size=n) * 0.3) > 0)
    cols = [f"c{i}" for i in range(nf)]
    df = pd.DataFrame(X, columns=cols)
    df["label"] = y.astype(np.float32)
    model = LightGBMClassifier(featureCols=cols, numIterations=20,
                               numLeaves=15).fit(df)
    return model, df, cols


def test_tabular_shap_recovers_signal(model_and_data):
    model, df, cols = model_and_data
    shap = TabularSHAP(inputCols=cols, model=model, targetCol="probability",
                       targetClasses=[1], numSamples=256,
                       backgroundData=df.head(100))
    out = shap.transform(df.head(8))
    exp = np.stack(out["explanation"].to_numpy())  # (8, 1, m+1)
    assert exp.shape == (8, 1, 6)
    mean_abs = np.abs(exp[:, 0, 1:]).mean(axis=0)
    # informative features dominate (tree also uses noise features a little)
    assert mean_abs[0] > 1.5 * mean_abs[2:].max()
    assert mean_abs[1] > 1.5 * mean_abs[2:].max()


def test_tabular_shap_additivity(model_and_data):
    model, df, cols = model_and_data
    shap = TabularSHAP(inputCols=cols, model=model, targetCol="probability",
                       targetClasses=[1], numSamples=64,
                       backgroundData=df.head(50))
    rows = df.head(4)
    out = shap.transform(rows)
    exp = np.stack(out["explanation"].to_numpy())
    probs = np.stack(model.transform(rows)["probability"].to_numpy())[:, 1]
    # base + sum(phi) == model output (efficiency constraint)
    np.testing.assert_allclose(exp[:, 0, :].sum(axis=1), probs, atol=1e-3)


def test_vector_shap(model_and_data):
    model, df, cols = model_and_data
    X = df[cols].to_numpy(dtype=np.float32)
    vdf = pd.DataFrame({"features": matrix_to_vector_column(X)})
    vdf_bg = pd.DataFrame({"features": matrix_to_vector_column(X[:50])})
    vmodel = LightGBMClassifier(numIterations=10, numLeaves=7).fit(
        pd.DataFrame({"features": matrix_to_vector_column(X),
                      "label": df["label"]}))
    shap = VectorSHAP(model=vmodel, targetCol="probability", targetClasses=[1],
                      numSamples=64, backgroundData=vdf_bg)
    out = shap.transform(vdf.head(3))
    exp = np.stack(out["explanation"].to_numpy())
    assert exp.shape == (3, 1, 6)


def test_tabular_lime(model_and_data):
    model, df, cols = model_and_data
    lime = TabularLIME(inputCols=cols, model=model, targetCol="probability",
                       targetClasses=[1], numSamples=400,
                       backgroundData=df.head(100))
    out = lime.transform(df.head(5))
    exp = np.stack(out["explanation"].to_numpy())  # (5, 1, m)
    assert exp.shape == (5, 1, 5)
    mean_abs = np.abs(exp[:, 0, :]).mean(axis=0)
    assert mean_abs[:2].min() > mean_abs[2:].max()


class _TokenCountModel:
    """Toy text model: score = #occurrences of 'good'."""
    def transform(self, df):
        out = df.copy()
        out["score"] = [float(t.split().count("good")) for t in df["text"]]
        return out


def test_text_shap_and_lime():
    df = pd.DataFrame({"text": ["good movie bad plot good acting"]})
    shap = TextSHAP(model=_TokenCountModel(), targetCol="score",
                    targetClasses=[0], numSamples=128)
    out = shap.transform(df)
    exp = out["explanation"].iloc[0]
    toks = out["tokens"].iloc[0]
    contribs = exp[0, 1:]
    good_idx = [i for i, t in enumerate(toks) if t == "good"]
    bad_idx = [i for i, t in enumerate(toks) if t != "good"]
    assert min(contribs[good_idx]) > 0.5
    assert max(abs(contribs[bad_idx])) < 0.2

    lime = TextLIME(model=_TokenCountModel(), targetCol="score",
                    targetClasses=[0], numSamples=256)
    out2 = lime.transform(df)
    exp2 = out2["explanation"].iloc[0]
    assert exp2[0][good_idx].min() > exp2[0][bad_idx].max()


def test_image_superpixel_and_lime():
    from mmlspark_amd.explainers.lime import ImageLIME

    class _BrightModel:
        def transform(self, df):
            out = df.copy()
            out["score"] = [float(np.asarray(v)[:8, :8].mean()) for v in df["image"]]
            return out

    rng = np.random.default_rng(0)
    img = rng.integers(0, 40, size=(32, 32, 3)).astype(np.uint8)
    img[:8, :8] = 250  # bright corner drives the model
    df = pd.DataFrame({"image": [img]})
    lime = ImageLIME(model=_BrightModel(), targetCol="score", targetClasses=[0],
                     cellSize=8, numSamples=128)
    out = lime.transform(df)
    exp = out["explanation"].iloc[0][0]
    segs = out["superpixels"].iloc[0]
    corner_seg = segs[:8, :8].flatten()
    vals, counts = np.unique(corner_seg, return_counts=True)
    main_seg = int(vals[counts.argmax()])
    assert exp[main_seg] == exp.max()


def test_vector_lime(model_and_data):
    from mmlspark_amd.explainers.lime import VectorLIME
    model, df, cols = model_and_data
    X = df[cols].to_numpy(dtype=np.float32)
    vdf = pd.DataFrame({"features": matrix_to_vector_column(X)})
    vmodel = LightGBMClassifier(numIterations=10, numLeaves=7).fit(
        pd.DataFrame({"features": matrix_to_vector_column(X),
                      "label": df["label"]}))
    lime = VectorLIME(model=vmodel, targetCol="probability", targetClasses=[1],
                      numSamples=300,
                      backgroundData=vdf.head(80))
    out = lime.transform(vdf.head(3))
    exp = np.stack(out["explanation"].to_numpy())
    assert exp.shape == (3, 1, 5)
    mean_abs = np.abs(exp[:, 0, :]).mean(axis=0)
    assert mean_abs[:2].min() > mean_abs[2:].max()


def test_image_shap():
    from mmlspark_amd.explainers.shap import ImageSHAP

    class _CornerModel:
        def transform(self, df):
            out = df.copy()
            out["score"] = [float(np.asarray(v)[:8, :8].mean())
                            for v in df["image"]]
            return out

    rng = np.random.default_rng(1)
    img = rng.integers(0, 30, size=(24, 24, 3)).astype(np.uint8)
    img[:8, :8] = 240
    df = pd.DataFrame({"image": [img]})
    shap = ImageSHAP(model=_CornerModel(), targetCol="score",
                     targetClasses=[0], cellSize=8, numSamples=96)
    out = shap.transform(df)
    exp = out["explanation"].iloc[0][0]
    segs = out["superpixels"].iloc[0]
    vals, counts = np.unique(segs[:8, :8].flatten(), return_counts=True)
    main_seg = int(vals[counts.argmax()])
    # the bright-corner superpixel carries the largest contribution
    assert exp[1 + main_seg] == exp[1:].max()


def test_explainer_save_load_with_dataframe_param(tmp_path, model_and_data):
    """ComplexParam DataFrame side-file (parquet) round trip through an
    explainer — the ComplexParamsWriter path for backgroundData
    (serialize.ComplexParam parity)."""
    from mmlspark_amd.core.serialize import load_stage, save_stage
    model, df, cols = model_and_data
    shap = TabularSHAP(inputCols=cols, model=model, targetCol="probability",
                       targetClasses=[1], numSamples=64,
                       backgroundData=df.head(40))
    path = str(tmp_path / "shap_stage")
    save_stage(shap, path)
    back = load_stage(path)
    pd.testing.assert_frame_equal(back.get("backgroundData"),
                                  shap.get("backgroundData"))
    out = back.transform(df.head(2))
    assert np.stack(out["explanation"].to_numpy()).shape == (2, 1, 6)


def test_tabular_shap_generic_model_fallback():
    """A model WITHOUT score_matrix (any plain Transformer) must still be
    explainable — the DataFrame fallback of _score_matrix."""
    from mmlspark_amd.core.pipeline import Model

    class LinModel(Model):
        def _transform(self, df):
            out = df.copy()
            out["probability"] = [
                np.array([0.0, float(r["f0"] * 2 + r["f1"])])
                for _, r in df.iterrows()]
            return out

    rng = np.random.default_rng(0)
    df = pd.DataFrame({"f0": rng.normal(size=8), "f1": rng.normal(size=8),
                       "f2": rng.normal(size=8)})
    shap = TabularSHAP(inputCols=["f0", "f1", "f2"], model=LinModel(),
                       targetCol="probability", targetClasses=[1],
                       backgroundData=df, numSamples=1024, rowBatch=4)
    out = shap.transform(df)
    exp = np.stack(out[shap.get("outputCol")].to_numpy())
    assert exp.shape == (8, 1, 4)
    # linear model: phi_f0 ≈ 2*(x0 - E[x0]), phi_f2 ≈ 0
    x0 = df["f0"].to_numpy()
    np.testing.assert_allclose(exp[:, 0, 1], 2 * (x0 - x0.mean()),
                               atol=0.1)
    np.testing.assert_allclose(exp[:, 0, 3], 0, atol=0.1)


def test_lime_metrics_col_r2():
    """metricsCol emits the local surrogate's per-class R²
    (LIMEBase's r2 metrics output)."""
    from mmlspark_amd.models.gbdt.estimators import LightGBMClassifier
    rng = np.random.default_rng(2)
    X = rng.normal(size=(800, 5)).astype(np.float32)
    cols = [f"f{i}" for i in range(5)]
    df = pd.DataFrame(X, columns=cols)
    df["label"] = (X[:, 0] > 0).astype(np.float32)
    model = LightGBMClassifier(featureCols=cols, numIterations=20,
                               numLeaves=7).fit(df)
    lime = TabularLIME(inputCols=cols, model=model, targetCol="probability",
                       targetClasses=[1], backgroundData=df.head(100),
                       numSamples=400, metricsCol="r2")
    out = lime.transform(df.head(6))
    r2 = np.stack(out["r2"].to_numpy())
    assert r2.shape == (6, 1)
    assert (r2 > 0.2).all() and (r2 <= 1.0 + 1e-9).all(), r2
