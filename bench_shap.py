#!/usr/bin/env python3
"""KernelSHAP throughput (BASELINE config #5: TabularSHAP over a trained
LightGBM, 1k background x ~10k perturbations per row, batched through the
HIP forest-scoring kernel). value = explanations/sec."""
import argparse
import json
import os
import sys
import time

import numpy as np
import pandas as pd
import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rows", type=int, default=16, help="rows to explain")
    ap.add_argument("--features", type=int, default=20)
    ap.add_argument("--samples", type=int, default=10000)
    ap.add_argument("--background", type=int, default=1000)
    ap.add_argument("--trees", type=int, default=100)
    args = ap.parse_args()

    sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
    from mmlspark_amd.explainers.shap import TabularSHAP
    from mmlspark_amd.models.gbdt.estimators import LightGBMClassifier

    use_gpu = torch.cuda.is_available()
    if not use_gpu:
        args.samples = min(args.samples, 512)
        args.rows = min(args.rows, 4)

    rng = np.random.default_rng(0)
    n, nf = 50_000, args.features
    X = rng.normal(size=(n, nf)).astype(np.float32)
    w = rng.normal(size=nf)
    y = ((X @ w) + rng.normal(size=n) * 0.5 > 0).astype(np.float32)
    cols = [f"c{i}" for i in range(nf)]
    df = pd.DataFrame(X, columns=cols)
    df["label"] = y
    model = LightGBMClassifier(featureCols=cols, numIterations=args.trees,
                               numLeaves=31).fit(df)

    shap = TabularSHAP(inputCols=cols, model=model, targetCol="probability",
                       targetClasses=[1], numSamples=args.samples,
                       backgroundData=df.head(args.background),
                       rowBatch=args.rows)
    # warmup
    shap.transform(df.head(2))
    t0 = time.perf_counter()
    out = shap.transform(df.head(args.rows))
    elapsed = time.perf_counter() - t0
    exp = np.stack(out["explanation"].to_numpy())
    probs = np.stack(model.transform(df.head(args.rows))["probability"]
                     .to_numpy())[:, 1]
    additivity = float(np.abs(exp[:, 0, :].sum(axis=1) - probs).max())
    scored_rows = args.rows * (args.samples + args.background) + args.rows

    print(json.dumps({
        "metric": "kernel_shap_explanations_per_sec",
        "value": args.rows / elapsed,
        "unit": "explanations/s", "n_gpus": 1 if use_gpu else 0,
        "higher_is_better": True,
        "elapsed_s": elapsed,
        "model_scores_per_sec": scored_rows / elapsed,
        "additivity_max_err": additivity,
        "dtype": "fp32", "data": "synthetic",
        "config": {"rows": args.rows, "features": nf,
                   "perturbations_per_row": args.samples,
                   "background": args.background, "trees": args.trees},
    }), flush=True)


if __name__ == "__main__":
    main()
