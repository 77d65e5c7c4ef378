"""FindBestModel rocCurve/scoredDataset params (FindBestModel.scala:134)."""
import numpy as np
import pandas as pd



def test_best_model_roc_and_scored_dataset():
    from mmlspark_amd.stages.automl import FindBestModel
    from mmlspark_amd.models.gbdt.estimators import LightGBMClassifier
    rng = np.random.default_rng(0)
    X = rng.normal(size=(400, 6)).astype(np.float32)
    y = (X[:, 0] > 0).astype(np.float64)
    df = pd.DataFrame({"features": list(X), "label": y})
    models = [LightGBMClassifier(numIterations=5, numLeaves=7).fit(df),
              LightGBMClassifier(numIterations=2, numLeaves=3).fit(df)]
    best = FindBestModel(models=models, evaluationMetric="AUC").fit(df)
    roc = best.getRocCurve()
    assert roc is not None and {"falsePositiveRate",
                                "truePositiveRate"} <= set(roc.columns)
    assert roc["truePositiveRate"].iloc[-1] == 1.0
    scored = best.getScoredDataset()
    assert scored is not None and "probability" in scored.columns


def test_tuned_model_save_load_round_trip(tmp_path):
    """TuneHyperparametersModel persists its nested fitted best model
    (ComplexParam stage serializer) and scores identically after reload."""
    from mmlspark_amd.core.serialize import load_stage
    from mmlspark_amd.models.gbdt.estimators import LightGBMClassifier
    from mmlspark_amd.stages.automl import (HyperparamBuilder,
                                            TuneHyperparameters)
    rng = np.random.default_rng(0)
    X = rng.normal(size=(300, 4)).astype(np.float32)
    y = (X[:, 0] > 0).astype(np.float64)
    df = pd.DataFrame({"features": list(X), "label": y})
    sp = HyperparamBuilder().addDiscrete("numLeaves", [7, 15]).build()
    m = TuneHyperparameters(models=[LightGBMClassifier(numIterations=3)],
                            paramSpace=sp, numRuns=2, numFolds=2).fit(df)
    p = str(tmp_path / "m")
    m.save(p)
    m2 = load_stage(p)
    assert (m.transform(df)["prediction"]
            == m2.transform(df)["prediction"]).all()
    assert m2.get("bestMetric") == m.get("bestMetric")
