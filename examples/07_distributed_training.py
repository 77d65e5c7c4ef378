"""Distributed GBDT training — the torchrun rank model end to end.

Run on one GPU / CPU:        python examples/07_distributed_training.py
Run on all GPUs of a node:   python -m torch.distributed.run --nnodes=1 \\
    --nproc-per-node 8 --master-addr 127.0.0.1 examples/07_distributed_training.py

Each rank generates (in practice: loads) ITS OWN row shard; `fit` detects
the process group and synchronizes histograms over RCCL/xGMI (gloo on CPU).
The fitted model is identical on every rank, bit for bit — rank 0 persists
it (docs/distributed.md, docs/migration.md)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import pandas as pd
import torch

from mmlspark_amd.models.gbdt.estimators import LightGBMClassifier
from mmlspark_amd.parallel.comm import init_from_env

comm = init_from_env()
rank, world = comm.rank, comm.world_size
if torch.cuda.is_available():
    torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", rank)))

# per-rank shard of a shared underlying distribution
rng = np.random.default_rng(1000 + rank)
n, nf = 200_000, 20
X = rng.normal(size=(n, nf)).astype(np.float32)
w = np.random.default_rng(7).normal(size=nf)  # same signal on every rank
y = ((X @ w + rng.normal(size=n) * 0.5) > 0).astype(np.float32)
df = pd.DataFrame({"features": list(X), "label": y})

model = LightGBMClassifier(
    numIterations=30, numLeaves=31, learningRate=0.2,
    device="cuda" if torch.cuda.is_available() else "cpu").fit(df)

scored = model.transform(df.head(10_000))
acc = float((scored["prediction"].to_numpy() == y[:10_000]).mean())
print(f"[rank {rank}/{world}] local holdout acc {acc:.3f}, "
      f"{model.booster.num_trees} trees")

if rank == 0:
    model.save("/tmp/distributed_gbdt_model")
    print("rank 0 saved /tmp/distributed_gbdt_model "
          "(identical on every rank)")
