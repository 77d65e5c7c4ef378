"""Committed accuracy-benchmark regression (Benchmarks-trait parity):
metric values on fixed synthetic data must not regress below the committed
CSV within its precision (lightgbm benchmarks_VerifyLightGBMClassifier.csv
style — AUC per boosting type, VW MSE)."""
import os

import numpy as np
import pandas as pd
import pytest

from mmlspark_amd.utils.benchmarks import BenchmarkRunner

RESOURCE_DIR = os.path.join(os.path.dirname(__file__), "resources",
                            "benchmarks")


def _fixed_binary(n=3000, nf=10, seed=42):
    rng = np.random.default_rng(seed)
    X = rng.normal(size=(n, nf)).astype(np.float32)
    w = rng.normal(size=nf)
    y = ((X @ w + rng.normal(size=n) * 0.5) > 0).astype(np.float32)
    return pd.DataFrame({"features": list(X), "label": y})


def test_lightgbm_benchmark_csv():
    from sklearn.metrics import roc_auc_score

    from mmlspark_amd.models.gbdt.estimators import LightGBMClassifier
    df = _fixed_binary()
    runner = BenchmarkRunner("VerifyLightGBMClassifier", RESOURCE_DIR)
    for boosting in ("gbdt", "rf", "dart", "goss"):
        m = LightGBMClassifier(numIterations=20, numLeaves=15, seed=0,
                               learningRate=0.2, boostingType=boosting,
                               baggingFraction=0.8, baggingFreq=1).fit(df)
        prob = np.stack(m.transform(df)["probability"].to_numpy())[:, 1]
        auc = roc_auc_score(df["label"], prob)
        # reference commits AUC with precision 0.07
        # (benchmarks_VerifyLightGBMClassifier.csv)
        runner.add(f"auc_synthetic_{boosting}", auc, precision=0.07)
    problems = runner.compare()
    assert not problems, problems


def test_vw_benchmark_csv():
    from mmlspark_amd.core.schema import SparseVector
    from mmlspark_amd.models.vw.estimators import VowpalWabbitRegressor
    rng = np.random.default_rng(7)
    size = 1 << 14
    w_true = rng.normal(size=size) * 0.05
    rows, ys = [], []
    for _ in range(3000):
        idx = np.unique(rng.integers(0, size, size=25))
        val = np.ones(len(idx), dtype=np.float32)
        rows.append(SparseVector(size, idx.astype(np.int32), val))
        ys.append(float((w_true[idx]).sum()))
    df = pd.DataFrame({"features": rows, "label": ys})
    m = VowpalWabbitRegressor(numPasses=8, numBits=14, learningRate=0.3,
                              holdoutOff=True).fit(df)
    pred = m.transform(df)["prediction"].to_numpy()
    mse = float(((pred - np.asarray(ys)) ** 2).mean())
    runner = BenchmarkRunner("VerifyVowpalWabbitRegressor", RESOURCE_DIR)
    runner.add("mse_synthetic_default", mse, precision=0.02,
               higher_is_better=False)
    problems = runner.compare()
    assert not problems, problems


def test_train_classifier_benchmark_csv():
    """TrainClassifier AUROC/AUPR bars (benchmarks_VerifyTrainClassifier.csv
    parity: PimaIndian-shaped 8-feature binary data, auto-featurized)."""
    from mmlspark_amd.stages.train import (ComputeModelStatistics,
                                           TrainClassifier)
    from mmlspark_amd.models.gbdt.estimators import LightGBMClassifier

    rng = np.random.default_rng(7)
    n = 768  # PimaIndian size
    X = rng.normal(size=(n, 8)).astype(np.float32)
    w = rng.normal(size=8)
    y = ((X @ w + rng.normal(size=n) * 1.5) > 0).astype(np.float32)
    df = pd.DataFrame({f"f{i}": X[:, i] for i in range(8)})
    df["label"] = y
    model = TrainClassifier(model=LightGBMClassifier(numIterations=20,
                                                     numLeaves=7, seed=0),
                            labelCol="label").fit(df)
    stats = ComputeModelStatistics(labelCol="label").transform(
        model.transform(df))
    runner = BenchmarkRunner("VerifyTrainClassifier", RESOURCE_DIR)
    runner.add("auroc_pima_shaped", float(stats.iloc[0]["AUC"]),
               precision=0.05)
    runner.add("aupr_pima_shaped", float(stats.iloc[0].get(
        "AUPR", stats.iloc[0].get("precision", 0.0))), precision=0.08)
    problems = runner.compare()
    assert not problems, problems


def test_vw_multipass_quadratic_benchmark_csv():
    """committed MSE bar for a '--passes 4 -q ::' style
    run — multi-pass replay + raw-input namespace crossing
    (benchmarks_VerifyVowpalWabbitRegressor.csv discipline)."""
    from mmlspark_amd.models.vw.estimators import VowpalWabbitRegressor
    rng = np.random.default_rng(13)
    n = 3000
    xa = rng.normal(size=(n, 4)).astype(np.float32)
    xb = rng.normal(size=(n, 4)).astype(np.float32)
    # label needs the a×b cross terms: linear-only MSE stays high
    y = (xa[:, 0] * xb[:, 0] + 0.5 * xa[:, 1] * xb[:, 1]
         + 0.1 * rng.normal(size=n)).astype(np.float32)
    df = pd.DataFrame({"afeat": list(xa), "bfeat": list(xb), "label": y})
    m = VowpalWabbitRegressor(featuresCol="afeat",
                              additionalFeatures=["bfeat"],
                              passThroughArgs="--passes 16 -q ::",
                              numBits=18, learningRate=0.02,
                              adaptive=True, holdoutOff=True).fit(df)
    pred = m.transform(df)["prediction"].to_numpy()
    mse = float(((pred - y) ** 2).mean())
    runner = BenchmarkRunner("VerifyVowpalWabbitRegressor", RESOURCE_DIR)
    runner.add("mse_synthetic_passes4_quadratic", mse, precision=0.05,
               higher_is_better=False)
    problems = runner.compare()
    assert not problems, problems
