"""Kitchen-sink integration: messy mixed-type data → cleaning → auto-
featurization → GBDT training inside a Pipeline → persistence round trip →
metrics → SHAP column → serving the LOADED pipeline over HTTP.  One flow a
migrating MMLSpark user would actually run."""
import json

import numpy as np
import pandas as pd

from mmlspark_amd.core.pipeline import Pipeline, PipelineModel
from mmlspark_amd.core.serialize import load_stage
from mmlspark_amd.models.gbdt.estimators import LightGBMClassifier
from mmlspark_amd.serving.server import ServingServer
from mmlspark_amd.stages.featurize import CleanMissingData, Featurize
from mmlspark_amd.stages.train import ComputeModelStatistics


def _data(n=4000, seed=0):
    rng = np.random.default_rng(seed)
    df = pd.DataFrame({
        "age": rng.normal(40, 12, n).round(),
        "hours": rng.normal(38, 9, n),
        "city": rng.choice(["tokyo", "lima", "oslo", "pune"], n),
        "member": rng.choice([True, False], n),
    })
    logit = ((df.age - 40) / 12 + (df.hours - 38) / 9
             + df.city.map({"tokyo": 1.0, "lima": -1.0, "oslo": 0.3,
                            "pune": -0.3}) + df.member * 0.8)
    df["label"] = (logit + rng.normal(0, 0.7, n) > 0).astype(np.float32)
    # inject missing values the pipeline must clean
    df.loc[rng.choice(n, n // 20, replace=False), "age"] = np.nan
    return df


def test_full_pipeline_save_load_score_explain_serve(tmp_path):
    df = _data()
    pipe = Pipeline(stages=[
        CleanMissingData(inputCols=["age"], outputCols=["age"],
                         cleaningMode="Median"),
        Featurize(inputCols=["age", "hours", "city", "member"],
                  outputCol="features"),
        LightGBMClassifier(numIterations=25, numLeaves=15, learningRate=0.2,
                           featuresShapCol="shap"),
    ])
    model = pipe.fit(df)
    assert isinstance(model, PipelineModel)

    scored = model.transform(df)
    stats = ComputeModelStatistics(labelCol="label").transform(scored)
    auc = float(stats.iloc[0]["AUC"])
    assert auc > 0.85, auc

    # SHAP additivity survives the full pipeline
    shap = np.stack(scored["shap"].to_numpy())
    raw = np.stack(scored["rawPrediction"].to_numpy())[:, 1]
    assert np.abs(shap.sum(axis=1) - raw).max() < 1e-3

    # persistence round trip scores identically
    path = str(tmp_path / "pipeline_model")
    model.save(path)
    back = load_stage(path)
    rescored = back.transform(df.head(100))
    np.testing.assert_allclose(
        np.stack(rescored["probability"].to_numpy()),
        np.stack(scored.head(100)["probability"].to_numpy()), rtol=1e-6)

    # serve the LOADED pipeline over HTTP (continuous mode)
    def handler(payloads):
        rows = pd.DataFrame(payloads)
        out = back.transform(rows)
        return [{"p": float(p[1])} for p in out["probability"]]

    srv = ServingServer(handler, port=0, mode="continuous").start()
    try:
        import http.client
        conn = http.client.HTTPConnection("127.0.0.1", srv.port)
        row = df.iloc[0][["age", "hours", "city", "member"]].to_dict()
        row = {k: (v.item() if hasattr(v, "item") else v)
               for k, v in row.items()}
        conn.request("POST", "/", json.dumps(row).encode(),
                     {"Content-Type": "application/json"})
        resp = json.loads(conn.getresponse().read())
        assert 0.0 <= resp["p"] <= 1.0
        expect = float(scored.iloc[0]["probability"][1])
        assert abs(resp["p"] - expect) < 1e-5
    finally:
        srv.stop()
