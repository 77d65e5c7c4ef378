"""End-to-end tabular classification: featurize mixed columns, train a GBDT,
evaluate, explain, and save — the core MMLSpark workflow
(adult-census-shaped synthetic data; runs on CPU or MI355X)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import pandas as pd

from mmlspark_amd.explainers.shap import TabularSHAP
from mmlspark_amd.models.gbdt.estimators import LightGBMClassifier
from mmlspark_amd.stages.train import ComputeModelStatistics, TrainClassifier

rng = np.random.default_rng(0)
n = 32_000
df = pd.DataFrame({
    "age": rng.integers(17, 90, n).astype(float),
    "education": rng.choice(["hs", "college", "masters", "phd"], n),
    "hours_per_week": rng.normal(40, 12, n).clip(1, 99),
    "capital_gain": rng.exponential(500, n),
})
logit = (df.age - 38) / 10 + (df.education == "masters") * 1.2 \
    + (df.education == "phd") * 2.0 + (df.hours_per_week - 40) / 20
df["label"] = (logit + rng.normal(0, 1, n) > 0.5).astype(int)

# auto-featurize + fit (TrainClassifier wraps featurization + label indexing)
model = TrainClassifier(model=LightGBMClassifier(numIterations=100,
                                                 numLeaves=31)).fit(df)
scored = model.transform(df)
stats = ComputeModelStatistics(evaluationMetric="classification").transform(scored)
print(stats[["accuracy", "AUC"]].to_string(index=False))

# SHAP explanations for a few rows through the same scoring path
inner = model.get("innerModel")
feats = model.get("featurizerModel").transform(df)
cols = ["age", "hours_per_week", "capital_gain"]
shap = TabularSHAP(inputCols=cols, model=LightGBMClassifier(
    featureCols=cols, numIterations=50).fit(df.assign(label=df.label)),
    targetCol="probability", targetClasses=[1], numSamples=256,
    backgroundData=df.head(200))
print(shap.transform(df.head(2))["explanation"].iloc[0])

model.save("/tmp/census_model")
print("saved to /tmp/census_model")
