#!/usr/bin/env python3
"""Histogram-kernel microbenchmark: sweep GPB (features-per-block) and leaf
size on the GPU; reports us/call and effective GB/s."""
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def run():
    from mmlspark_amd.ops import backend
    assert torch.cuda.is_available()
    g = torch.Generator().manual_seed(0)
    n, nf, nb = 10_000_000, 100, 255
    ng = (nf + 3) // 4
    binned = torch.randint(0, nb, (ng, n, 4), generator=g,
                           dtype=torch.uint8).cuda()
    grad = torch.randn(n, generator=g).cuda()
    hess = torch.rand(n, generator=g).cuda()
    results = {}
    for gpb in (1, 2, 4, 8):
        os.environ["MMLSPARK_HIST_GPB"] = str(gpb)
        row = {}
        for frac in (1.0, 0.25, 0.05):
            m = int(n * frac)
            rows = torch.randperm(n, generator=g)[:m].to(torch.int32)\
                .sort().values.cuda()
            # warmup
            for _ in range(2):
                backend.hist_build(binned, rows, grad, hess, nb)
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            iters = 10 if frac < 0.5 else 5
            for _ in range(iters):
                backend.hist_build(binned, rows, grad, hess, nb)
            torch.cuda.synchronize()
            us = (time.perf_counter() - t0) / iters * 1e6
            nblocks = (ng + gpb - 1) // gpb
            bytes_moved = m * (4 + 8 * nblocks + ng * 4)  # rows + g/h per fblock + binned
            row[f"m={m}"] = {"us": round(us, 1),
                             "GB/s": round(bytes_moved / us / 1e3, 1)}
        results[f"GPB={gpb}"] = row
        print(f"GPB={gpb}: {json.dumps(row)}", flush=True)
    return results


if __name__ == "__main__":
    run()
