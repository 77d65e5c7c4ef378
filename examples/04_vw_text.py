"""VW-style text classification: hash featurize raw columns, sparse SGD."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import pandas as pd

from mmlspark_amd.models.vw.estimators import VowpalWabbitClassifier
from mmlspark_amd.models.vw.featurizer import VowpalWabbitFeaturizer

rng = np.random.default_rng(0)
pos = ["great movie", "loved it", "fantastic acting", "great plot twist"]
neg = ["terrible movie", "hated it", "awful acting", "boring plot"]
texts = [rng.choice(pos if i % 2 == 0 else neg) for i in range(4000)]
df = pd.DataFrame({"text": texts, "label": [i % 2 == 0 for i in range(4000)]})
df["label"] = df.label.astype(float)

feats = VowpalWabbitFeaturizer(stringSplitInputCols=["text"], numBits=18)
dff = feats.transform(df)
model = VowpalWabbitClassifier(numPasses=3, normalized=True).fit(dff)
out = model.transform(dff)
print("train acc:", (out.prediction == df.label).mean())
