"""Quantile binning — builds the per-feature bin mapper shared by all ranks.

Equivalent of the dataset construction the reference delegates to
``LGBM_DatasetCreateFromMat`` (dataset/DatasetAggregator.scala:335): quantile
boundaries from a row sample, then the full matrix binned to uint8 in the
feature-interleaved (ngroups, n_rows, 4) device layout consumed by the HIP
histogram kernel.  Distributed: every rank contributes a fixed-size sample
(all_gather), then computes identical boundaries — no broadcast needed.
"""
from __future__ import annotations

from typing import Optional

import torch

from ...ops import backend
from ...parallel.comm import Comm


class BinMapper:
    """Per-feature quantile bin boundaries. upper_bounds[f, b] = largest value
    in bin b.  Categorical features (categoricalSlotIndexes parity) are binned
    by category id: bin = clamp(round(x), 0, n_bins-1)."""

    def __init__(self, upper_bounds: torch.Tensor, n_bins: int,
                 categorical: Optional[list] = None):
        self.upper_bounds = upper_bounds  # (nf, n_bins-1) float32, +inf padded
        self.n_bins = n_bins
        self.n_features = upper_bounds.shape[0]
        self.categorical = sorted(set(categorical or []))

    @staticmethod
    def fit(X: torch.Tensor, n_bins: int = 255, sample_size: int = 200_000,
            comm: Optional[Comm] = None, seed: int = 0,
            categorical: Optional[list] = None,
            max_bin_by_feature: Optional[list] = None) -> "BinMapper":
        n, nf = X.shape
        gen = torch.Generator(device="cpu").manual_seed(seed)
        quota = max(1, sample_size // max(1, comm.world_size if comm else 1))
        k = min(n, quota)
        idx = torch.randperm(n, generator=gen)[:k].to(X.device)
        sample = X[idx]
        if comm is not None and comm.is_distributed:
            # all_gather needs EQUAL shapes on every rank.  Agree on a common
            # per-rank sample size first (one tiny collective): the largest
            # shard's k, capped at the quota.  Smaller shards pad by
            # repeating their sample (slight weight skew toward small shards
            # only — exact repeats do not move quantiles of a shard's own
            # distribution); an empty shard ships zeros
            ks = comm.all_gather(torch.tensor([k], dtype=torch.int64,
                                              device=X.device))
            common = int(min(quota, max(int(t[0]) for t in ks)))
            common = max(1, common)
            if sample.shape[0] == 0:
                sample = X.new_zeros((common, nf))
            elif sample.shape[0] < common:
                reps = (common + sample.shape[0] - 1) // sample.shape[0]
                sample = sample.repeat(reps, 1)[:common]
            sample = torch.cat(comm.all_gather(sample.contiguous()), dim=0)
        qs = torch.linspace(0, 1, n_bins, device=sample.device)[1:]  # n_bins-1 cuts
        sample = torch.nan_to_num(sample, nan=float("inf"))
        ub = torch.quantile(sample.double(), qs.double(), dim=0).t().float()  # (nf, n_bins-1)
        # strictly increasing boundaries; collapse duplicated quantiles
        ub = torch.cummax(ub, dim=1).values
        ub[:, -1] = float("inf")
        if max_bin_by_feature:  # maxBinByFeature: per-feature bin caps —
            # repeat the capped feature's quantile grid so only `cap` distinct
            # boundaries survive (duplicates collapse to one effective bin)
            for f, cap in enumerate(max_bin_by_feature[: ub.shape[0]]):
                if cap and 1 < cap < n_bins:
                    src = torch.linspace(0, ub.shape[1] - 1, cap - 1).long()
                    ub[f] = ub[f, src].repeat_interleave(
                        (ub.shape[1] + cap - 2) // (cap - 1))[: ub.shape[1]]
                    ub[f] = torch.cummax(ub[f], dim=0).values
                    ub[f, -1] = float("inf")
        return BinMapper(ub.contiguous(), n_bins, categorical)

    def transform(self, X: torch.Tensor) -> torch.Tensor:
        """(n, nf) float -> (ngroups, n, 4) uint8 feature-interleaved bins."""
        out = backend.bin_matrix(X, self.upper_bounds.to(X.device), self.n_bins)
        for f in self.categorical:  # category id IS the bin
            bins = torch.nan_to_num(X[:, f], nan=0.0).round().clamp(
                0, self.n_bins - 1).to(torch.uint8)
            out[f // 4, :, f % 4] = bins
        return out

    def bin_upper_value(self, feature: int, b: int) -> float:
        """Raw-value threshold for 'bin <= b' splits (used at predict time)."""
        if b >= self.n_bins - 1:
            return float("inf")
        return float(self.upper_bounds[feature, b])

    def state_dict(self):
        return {"upper_bounds": self.upper_bounds.cpu(), "n_bins": self.n_bins,
                "categorical": list(self.categorical)}

    @staticmethod
    def from_state(state):
        return BinMapper(state["upper_bounds"], int(state["n_bins"]),
                         state.get("categorical"))
