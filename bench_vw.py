#!/usr/bin/env python3
"""VW-equivalent throughput benchmark (BASELINE config #3:
VowpalWabbitClassifier, hashed 2^18-bit sparse text-shaped data,
sparse-SGD with per-pass weight all_reduce). One step = one pass over the
local shard; value = rows/sec aggregate."""
import argparse
import json
import os
import sys
import time

import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=5)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--rows", type=int, default=2_000_000)
    ap.add_argument("--nnz", type=int, default=50)
    ap.add_argument("--bits", type=int, default=18)
    args = ap.parse_args()

    sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
    from mmlspark_amd.ops import backend
    from mmlspark_amd.parallel.comm import init_from_env

    comm = init_from_env()
    rank, world = comm.rank, comm.world_size
    use_gpu = torch.cuda.is_available()
    device = torch.device("cuda") if use_gpu else torch.device("cpu")
    if use_gpu and world > 1:
        # modulo lets a one-GPU rehearsal run the multi-proc path (gloo)
        li = int(os.environ.get("LOCAL_RANK", rank)) % torch.cuda.device_count()
        torch.cuda.set_device(li)
        device = torch.device("cuda", li)

    n = args.rows if use_gpu else min(args.rows, 100_000)
    tbl = 1 << args.bits
    g = torch.Generator().manual_seed(99 + rank)
    idx = torch.randint(0, tbl, (n * args.nnz,), generator=g,
                        dtype=torch.int32).to(device)
    val = (torch.rand(n * args.nnz, generator=g) + 0.5).to(device)
    off = (torch.arange(n + 1, dtype=torch.int64) * args.nnz).to(device)
    w_true = torch.randn(tbl, generator=g)
    margins = torch.zeros(n)
    for s in range(0, n, 1 << 18):
        e = min(s + (1 << 18), n)
        sl = slice(int(s) * args.nnz, int(e) * args.nnz)
        seg = torch.repeat_interleave(torch.arange(e - s), args.nnz)
        margins[s:e] = torch.zeros(e - s).index_add_(
            0, seg, w_true[idx[sl].long().cpu()] * val[sl].cpu())
    labels = torch.sign(margins + torch.randn(n, generator=g) * 0.1).to(device)
    labels[labels == 0] = 1.0

    w = torch.zeros(tbl, dtype=torch.float32, device=device)
    gacc = torch.zeros(tbl, dtype=torch.float32, device=device)
    bs = 1 << 16

    def one_pass():
        for s in range(0, n, bs):
            e = min(s + bs, n)
            o = off[s:e + 1] - off[s]
            sl = slice(int(off[s]), int(off[e]))
            backend.vw_sgd_minibatch(idx[sl], val[sl], o, labels[s:e], w, gacc,
                                     0.5, 0.0, 0.5, "logistic")
        if comm.is_distributed:
            comm.all_reduce(w)
            w.div_(world)
            comm.all_reduce(gacc)
            gacc.div_(world)

    for _ in range(args.warmup):
        one_pass()
    comm.barrier()
    if use_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        one_pass()
    if use_gpu:
        torch.cuda.synchronize()
    comm.barrier()
    elapsed = time.perf_counter() - t0
    t = torch.tensor([elapsed], dtype=torch.float64)
    if comm.is_distributed:
        comm.all_reduce(t.to(device) if use_gpu else t, op="max")
    elapsed = float(t[0])

    if rank == 0:
        preds = backend.vw_predict(idx, val, off, w)
        acc = float((preds.sign() == labels).float().mean())
        print(json.dumps({
            "metric": "vw_train_rows_per_sec",
            "value": n * world * args.steps / elapsed,
            "unit": "rows/s", "n_gpus": world, "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1e3,
            "higher_is_better": True, "scaling": "weak",
            "vs_baseline": None, "dtype": "fp32", "data": "synthetic",
            "train_acc": acc,
            "config": {"model": "VowpalWabbitClassifier",
                       "rows_per_gpu": n, "nnz_per_row": args.nnz,
                       "bits": args.bits, "loss": "logistic",
                       "parallelism": f"dp{world}",
                       "sync": "per-pass RCCL all_reduce of 2^18 weights"},
        }), flush=True)


if __name__ == "__main__":
    main()
