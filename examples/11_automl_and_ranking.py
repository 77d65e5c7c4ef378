"""AutoML + model selection + ranking workflow: assemble features fast,
random-search hyperparameters with CV (TuneHyperparameters), pick the best
of several fitted models (FindBestModel, with ROC curve), and train/evaluate
a LambdaRank ranker — the reference's core/automl + ranking surface."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import pandas as pd

from mmlspark_amd.models.gbdt.estimators import (LightGBMClassifier,
                                                 LightGBMRanker)
from mmlspark_amd.stages.automl import (DiscreteHyperParam, FindBestModel,
                                        HyperparamBuilder, RangeHyperParam,
                                        TuneHyperparameters)
from mmlspark_amd.stages.featurize import FastVectorAssembler

rng = np.random.default_rng(7)
n = 4_000

# --- assemble scalar + vector columns into one features column ------------
df = pd.DataFrame({
    "f_scalar": rng.normal(size=n),
    "f_vec": list(rng.normal(size=(n, 6)).astype(np.float32)),
})
df = FastVectorAssembler(inputCols=["f_scalar", "f_vec"],
                         outputCol="features").transform(df)
X = np.stack(df["features"].to_numpy())
df["label"] = (X[:, 0] + X[:, 2] > 0).astype(np.float64)

# --- random-search CV over a hyperparameter space -------------------------
space = (HyperparamBuilder()
         .addHyperparam("numLeaves", DiscreteHyperParam([7, 15, 31]))
         .addHyperparam("learningRate", RangeHyperParam(0.05, 0.3))
         .build())
tuned = TuneHyperparameters(
    models=[LightGBMClassifier(numIterations=20)], paramSpace=space,
    numRuns=4, numFolds=3, parallelism=2, evaluationMetric="AUC").fit(df)
print("best config:", tuned.getBestModelInfo())

# --- FindBestModel across independently fitted candidates -----------------
candidates = [LightGBMClassifier(numIterations=it, numLeaves=15).fit(df)
              for it in (5, 40)]
best = FindBestModel(models=candidates, evaluationMetric="AUC").fit(df)
print(best.getEvaluationResults().to_string(index=False))
roc = best.getRocCurve()
print(f"ROC points: {len(roc)} (TPR reaches {roc['truePositiveRate'].iloc[-1]:.2f})")

# --- LambdaRank: grouped queries, NDCG early stopping ---------------------
nq, per_q = 120, 12
qdf = pd.DataFrame({
    "group": np.repeat(np.arange(nq), per_q),
    "features": list(rng.normal(size=(nq * per_q, 8)).astype(np.float32)),
})
rel = np.stack(qdf["features"].to_numpy())[:, 0] + rng.normal(
    0, .3, len(qdf))
qdf["label"] = pd.qcut(rel, 4, labels=False).astype(np.float64)
ranker = LightGBMRanker(numIterations=40, numLeaves=15, groupCol="group",
                        evalAt=[1, 3, 5]).fit(qdf)
scored = ranker.transform(qdf)
ndcg_in = scored.groupby("group")["prediction"].apply(
    lambda s: float(np.corrcoef(s, qdf.loc[s.index, "label"])[0, 1]))
print(f"mean per-query score/label correlation: {ndcg_in.mean():.3f}")
assert ndcg_in.mean() > 0.5
print("example 11 OK")
