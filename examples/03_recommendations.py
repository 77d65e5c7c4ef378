"""SAR recommendations: index raw ids, fit item-item similarity, recommend."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import pandas as pd

from mmlspark_amd.models.sar import SAR, RecommendationIndexer

rng = np.random.default_rng(0)
rows = []
for u in range(200):
    taste = u % 4
    for it in rng.choice(np.arange(25) + taste * 25, size=8, replace=False):
        rows.append({"user": f"u{u}", "item": f"i{it}", "rating": 1.0})
ratings = pd.DataFrame(rows)

indexer = RecommendationIndexer().fit(ratings)
indexed = indexer.transform(ratings)
sar = SAR(supportThreshold=2, similarityFunction="jaccard").fit(indexed)
recs = sar.recommendForAllUsers(5)
print(recs.head(3).to_string(index=False))
