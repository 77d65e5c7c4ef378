"""Probe: hist_build_fixed cost vs row-list shape (contiguous range gather
vs sorted random subset) to quantify the gather penalty that physical
data partitioning would remove.  Run on GPU: python tools/hist_gather_probe.py
"""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from mmlspark_amd.ops import backend


def bench(fn, iters=30):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us


def main():
    torch.manual_seed(0)
    n, nf, nb = 10_000_000, 100, 255
    ngroups = (nf + 3) // 4
    b8 = torch.randint(0, nb, (ngroups, 4, n), dtype=torch.uint8, device="cuda")
    binned = (b8[:, 0].to(torch.int32) | (b8[:, 1].to(torch.int32) << 8)
              | (b8[:, 2].to(torch.int32) << 16)
              | (b8[:, 3].to(torch.int32) << 24)).contiguous()
    grad = torch.randn(n, device="cuda")
    hess = torch.rand(n, device="cuda") + 0.5
    sg, sh = 2.0 ** 44 / n, 2.0 ** 24
    print(f"{'m':>10} {'contig us':>10} {'gather us':>10} {'ratio':>6}")
    for m in (16_384, 65_536, 262_144, 1_048_576, 4_194_304):
        rows_c = torch.arange(m, dtype=torch.int32, device="cuda")
        rows_g = torch.sort(torch.randperm(n, device="cuda")[:m])[0].to(
            torch.int32)
        t_c = bench(lambda: backend.hist_build_fixed(
            binned, rows_c, grad, hess, nb, sg, sh))
        t_g = bench(lambda: backend.hist_build_fixed(
            binned, rows_g, grad, hess, nb, sg, sh))
        print(f"{m:>10} {t_c:>10.1f} {t_g:>10.1f} {t_g / t_c:>6.2f}")


if __name__ == "__main__":
    main()
