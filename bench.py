#!/usr/bin/env python3
"""Flagship benchmark: distributed GBDT training throughput (BASELINE.json
config #2 — "LightGBMClassifier 10M rows x 100 numeric feats, RCCL histogram
all-reduce over xGMI").

One "step" = one boosting iteration over the full fixed synthetic matrix
(gradients + leaf-wise tree growth with histogram build/sync + prediction
update).  Weak scaling: each GPU rank holds its own 10M x 100 shard, so
value = rows/sec aggregated over all ranks = N * rows_per_gpu * steps / time.

Launch (driver contract):
  python bench.py --gpus 1 --steps K --warmup W                 # single rank
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W
"""
import argparse
import json
import os
import sys
import time

import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    # defaults sized so the no-flag run times ≥~1 s of steady-state steps
    # (round-1 verdict: 20 steps = 0.3 s was too short for the driver's
    # GPU-busy sampler to catch) while still finishing in well under a
    # minute end to end
    ap.add_argument("--steps", type=int, default=60)
    ap.add_argument("--warmup", type=int, default=10)
    ap.add_argument("--rows", type=int, default=10_000_000,
                    help="rows per GPU (weak scaling)")
    ap.add_argument("--features", type=int, default=100)
    ap.add_argument("--num-leaves", type=int, default=63)
    ap.add_argument("--categorical", type=int, default=0,
                    help="make the first K features categorical (ablation)")
    ap.add_argument("--sparse", action="store_true",
                    help="CSR ablation: train from sparse input")
    ap.add_argument("--nnz", type=int, default=20,
                    help="nonzeros per row in --sparse mode")
    ap.add_argument("--parallelism", default="data_parallel",
                    choices=["data_parallel", "voting_parallel"],
                    help="histogram sync strategy (voting reduces only the "
                         "globally-voted top-K features per split)")
    args = ap.parse_args()

    sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
    from mmlspark_amd.models.gbdt.objectives import make_objective
    from mmlspark_amd.models.gbdt.trainer import TrainConfig, TrainingSession
    from mmlspark_amd.parallel.comm import init_from_env

    comm = init_from_env()
    rank = comm.rank
    world = comm.world_size
    use_gpu = torch.cuda.is_available()
    device = torch.device("cuda") if use_gpu else torch.device("cpu")
    if use_gpu and world > 1:
        # modulo lets a one-GPU rehearsal run the multi-proc path (gloo)
        li = int(os.environ.get("LOCAL_RANK", rank)) % torch.cuda.device_count()
        torch.cuda.set_device(li)
        device = torch.device("cuda", li)

    n, nf = args.rows, args.features
    if not use_gpu:  # CPU smoke: shrink so the default invocation finishes fast
        n = min(n, 200_000)

    if args.sparse:
        # CSR generated DIRECTLY (a dense intermediate would need n×nf
        # floats — 1.5 TB at 4M×100k); label from a sparse linear model
        from mmlspark_amd.models.gbdt.sparse import CsrMatrix
        gen_dev = device if use_gpu else torch.device("cpu")
        if use_gpu:
            torch.cuda.manual_seed(1234 + rank)
        keep = torch.randint(0, nf, (n, args.nnz), device=gen_dev,
                             dtype=torch.int64).sort(dim=1).values
        val = torch.randn(n, args.nnz, device=gen_dev)
        w = torch.randn(nf, device=gen_dev) / (args.nnz ** 0.5)
        logits = (w[keep.reshape(-1)].view(n, args.nnz) * val).sum(dim=1)
        y = (logits + 0.5 * torch.randn(n, device=gen_dev) > 0).float()
        indptr = torch.arange(0, (n + 1) * args.nnz, args.nnz,
                              dtype=torch.int64, device=gen_dev)
        X = CsrMatrix(indptr, keep.to(torch.int32).reshape(-1),
                      val.reshape(-1), (n, nf))
        del keep, val, w, logits
        return run_session(args, comm, rank, world, use_gpu, device, X, y,
                           n, nf, cat_idx=None)

    # synthetic Higgs-like binary data, random-init (no network/datasets
    # here); generated directly on the device so an 8-rank launch does not
    # stage 8×4 GB through host RAM
    if use_gpu:
        torch.cuda.manual_seed(1234 + rank)
        X = torch.randn(n, nf, dtype=torch.float32, device=device)
        w = torch.randn(nf, device=device) / (nf ** 0.5)
        noise = 0.5 * torch.randn(n, device=device)
    else:
        g = torch.Generator(device="cpu").manual_seed(1234 + rank)
        X = torch.randn(n, nf, generator=g, dtype=torch.float32)
        w = torch.randn(nf, generator=g) / (nf ** 0.5)
        noise = 0.5 * torch.randn(n, generator=g)
    logits = X @ w + 0.3 * torch.sin(3 * X[:, 0]) + 0.3 * X[:, 1] * X[:, 2]
    y = (logits + noise > 0).float()
    cat_idx = None
    if args.categorical > 0:
        # ablation: first K features become 32-category ids whose effect on
        # the label is non-monotone (forces one-vs-rest set splits)
        k = min(args.categorical, nf)
        cats = torch.randint(0, 32, (n, k), device=X.device).float()
        parity = (cats.long() % 3 == 0).float().sum(dim=1)
        y = ((logits + noise + 0.5 * parity) > 0.5).float()
        X[:, :k] = cats
        cat_idx = list(range(k))
    X = X.to(device)
    y = y.to(device)
    return run_session(args, comm, rank, world, use_gpu, device, X, y, n,
                       nf, cat_idx)


def run_session(args, comm, rank, world, use_gpu, device, X, y, n, nf,
                cat_idx):
    from mmlspark_amd.models.gbdt.objectives import make_objective
    from mmlspark_amd.models.gbdt.trainer import TrainConfig, TrainingSession
    cfg = TrainConfig(num_iterations=args.warmup + args.steps,
                      num_leaves=args.num_leaves, learning_rate=0.1,
                      max_bin=255, min_data_in_leaf=20,
                      categorical_features=cat_idx,
                      parallelism=args.parallelism)
    objective = make_objective("binary")
    session = TrainingSession(X, y, cfg, objective, comm)

    for _ in range(args.warmup):
        session.step()

    comm.barrier()
    if use_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        session.step()
    if use_gpu:
        torch.cuda.synchronize()
    comm.barrier()
    elapsed = time.perf_counter() - t0

    # max over ranks
    t = torch.tensor([elapsed], dtype=torch.float64)
    if comm.is_distributed:
        t = t.to(device) if use_gpu else t
        comm.all_reduce(t, op="max")
    elapsed = float(t[0])

    if rank == 0:
        total_rows = n * world * args.steps
        rows_per_sec = total_rows / elapsed
        baseline = None  # reference publishes no absolute rows/sec
        print(json.dumps({
            "metric": "lightgbm_train_rows_per_sec",
            "value": rows_per_sec,
            "unit": "rows/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1e3,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": baseline,
            "dtype": "fp32",
            "data": "synthetic",
            "config": {
                "model": "LightGBMClassifier",
                "rows_per_gpu": n,
                "features": nf,
                "num_leaves": args.num_leaves,
                "max_bin": 255,
                "objective": "binary",
                "global_batch": n * world,
                "parallelism": (f"dp{world}" if args.parallelism ==
                                "data_parallel" else
                                f"voting{world} topK20"),
                "sync": (("RCCL" if use_gpu else
                          torch.distributed.get_backend())
                         + " histogram all_reduce"
                         + (" over xGMI" if use_gpu else "")) if world > 1
                        else "single rank",
                "categorical": args.categorical,
                "matrix": (f"csr nnz={args.nnz}" if args.sparse else "dense"),
            },
        }), flush=True)
        stats = session.stats.as_dict()
        print(f"# phase breakdown (rank0): {json.dumps(stats)}", file=sys.stderr)


if __name__ == "__main__":
    main()
