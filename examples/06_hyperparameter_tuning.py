"""AutoML: TuneHyperparameters random search over a GBDT + FindBestModel
(core/automl parity — TuneHyperparameters.scala:36, FindBestModel.scala:50),
then interop export: the winning model saved as stock LightGBM text and
re-loaded for scoring."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import pandas as pd

from mmlspark_amd.models.gbdt.estimators import (LightGBMClassificationModel,
                                                 LightGBMClassifier)
from mmlspark_amd.stages.automl import (FindBestModel, HyperparamBuilder,
                                        TuneHyperparameters)

rng = np.random.default_rng(0)
n, nf = 8000, 12
X = rng.normal(size=(n, nf)).astype(np.float32)
w = rng.normal(size=nf)
y = ((X @ w + 0.4 * np.sin(3 * X[:, 0]) + rng.normal(size=n) * 0.4) > 0)
df = pd.DataFrame({"features": list(X), "label": y.astype(np.float32)})

space = (HyperparamBuilder()
         .addRange("numLeaves", 7, 63)
         .addRange("learningRate", 0.05, 0.3)
         .addDiscrete("numIterations", [20, 40])
         .build())
tuner = TuneHyperparameters(models=[LightGBMClassifier()],
                            paramSpace=space, numRuns=6, numFolds=3,
                            evaluationMetric="AUC", seed=7, searchMode="random")
tuned = tuner.fit(df)
print("best params:", tuned.getBestModelInfo())

candidates = [LightGBMClassifier(numLeaves=nl, numIterations=25).fit(df)
              for nl in (7, 31)]
best = FindBestModel(models=candidates, evaluationMetric="AUC").fit(df)
print("FindBestModel AUC:", round(best.getBestModelMetrics(), 4))

# interop: export the winner as stock LightGBM native text + reload
txt = best.getBestModel().booster.to_lightgbm_text()
again = LightGBMClassificationModel.loadNativeModelFromString(txt)
scored = again.transform(df.head(5))
print(scored[["prediction"]].to_string(index=False))
