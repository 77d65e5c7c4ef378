"""Perturbation samplers per modality + KernelSHAP coalition generator.

Parity: core/.../explainers/Sampler.scala:28-230 (image-superpixel,
text-token, vector, tabular samplers) and KernelSHAPSampler.scala:44-129
(coalition sampling with exact enumeration for small sizes).
"""
from __future__ import annotations

import math
from typing import List, Tuple

import numpy as np


# --------------------------------------------------------------- coalitions
def kernel_shap_weight(m: int, s: int) -> float:
    """Shapley kernel weight for a coalition of size s out of m features."""
    if s == 0 or s == m:
        return 1e6  # handled exactly via v_null / v_full constraints
    return (m - 1) / (math.comb(m, s) * s * (m - s))


def sample_coalitions(m: int, n_samples: int, rng: np.random.Generator
                      ) -> Tuple[np.ndarray, np.ndarray]:
    """Coalition 0/1 matrix (n, m) + kernel weights, following the
    reference's strategy: enumerate |z| levels outward-in while the budget
    covers a full level (KernelSHAPSampler.scala:44-128), then sample the
    remaining budget from the Shapley-kernel size distribution."""
    rows: List[np.ndarray] = []
    weights: List[float] = []
    budget = n_samples
    sizes = list(range(1, m))
    # pair sizes (s, m-s) from the outside in
    level_order = []
    for s in range(1, m // 2 + 1):
        pair = [s] if s == m - s else [s, m - s]
        level_order.append(pair)
    remaining_sizes = []
    for pair in level_order:
        count = sum(math.comb(m, s) for s in pair)
        if count <= budget - len(remaining_sizes):
            for s in pair:
                for mask_idx in _enumerate_masks(m, s):
                    rows.append(mask_idx)
                    weights.append(kernel_shap_weight(m, s))
            budget -= count
        else:
            remaining_sizes.extend(pair)
    if remaining_sizes and budget > 0:
        probs = np.array([kernel_shap_weight(m, s) * math.comb(m, s)
                          for s in remaining_sizes])
        probs = probs / probs.sum()
        for _ in range(budget):
            s = int(rng.choice(remaining_sizes, p=probs))
            mask = np.zeros(m, dtype=np.float32)
            mask[rng.choice(m, size=s, replace=False)] = 1
            rows.append(mask)
            weights.append(kernel_shap_weight(m, s))
    if not rows:  # tiny m: all-ones fallback
        rows.append(np.ones(m, dtype=np.float32))
        weights.append(1.0)
    return np.stack(rows).astype(np.float32), np.asarray(weights, np.float64)


def _enumerate_masks(m: int, s: int):
    from itertools import combinations
    for comb in combinations(range(m), s):
        mask = np.zeros(m, dtype=np.float32)
        mask[list(comb)] = 1
        yield mask


def lime_sample_states(m: int, n_samples: int, rng: np.random.Generator
                       ) -> np.ndarray:
    """Bernoulli(0.5) on/off states for LIME."""
    return (rng.random((n_samples, m)) < 0.5).astype(np.float32)


# ----------------------------------------------------------------- modality
class VectorSampler:
    """Mask features of a dense vector against a background distribution."""

    def __init__(self, background: np.ndarray, rng: np.random.Generator):
        self.background = background  # (b, d)
        self.rng = rng

    def apply(self, instance: np.ndarray, states: np.ndarray,
              draws_per_state: int = 1) -> np.ndarray:
        """With draws_per_state == len(background) (the mean-aggregation
        path using the FULL background), cycle every background row once
        per coalition — exact marginalization instead of sampling noise."""
        n = states.shape[0]
        if (draws_per_state > 1
                and draws_per_state == len(self.background)
                and n % draws_per_state == 0):
            bidx = np.tile(np.arange(len(self.background)),
                           n // draws_per_state)
        else:
            bidx = self.rng.integers(0, len(self.background), size=n)
        bg = self.background[bidx]
        return states * instance[None, :] + (1 - states) * bg


class TabularSampler(VectorSampler):
    pass


class TextSampler:
    """Token on/off masking; off tokens removed."""

    def __init__(self, tokens: List[str]):
        self.tokens = tokens

    def apply(self, states: np.ndarray) -> List[str]:
        out = []
        for row in states:
            out.append(" ".join(t for t, on in zip(self.tokens, row) if on))
        return out


class ImageSampler:
    """Superpixel on/off masking; off segments filled with a constant."""

    def __init__(self, image: np.ndarray, segments: np.ndarray,
                 fill_value: float = 0.0):
        self.image = image
        self.segments = segments
        self.fill = fill_value
        self.n_segments = int(segments.max()) + 1

    def apply(self, states: np.ndarray) -> List[np.ndarray]:
        outs = []
        for row in states:
            img = self.image.copy()
            off = np.where(row < 0.5)[0]
            if len(off):
                mask = np.isin(self.segments, off)
                img[mask] = self.fill
            outs.append(img)
        return outs


def slic_superpixels(image: np.ndarray, cell_size: int = 16,
                     modifier: float = 10.0, n_iter: int = 5) -> np.ndarray:
    """SLIC-style superpixel clustering (Superpixel.scala:148 parity):
    k-means over (r,g,b,x,y) with grid-seeded centers."""
    h, w = image.shape[:2]
    img = image.astype(np.float32)
    if img.ndim == 2:
        img = img[:, :, None].repeat(3, axis=2)
    gy = np.arange(cell_size // 2, h, cell_size)
    gx = np.arange(cell_size // 2, w, cell_size)
    centers = np.array([[y, x] for y in gy for x in gx], dtype=np.float32)
    k = len(centers)
    yy, xx = np.mgrid[0:h, 0:w].astype(np.float32)
    feats = np.concatenate([img / modifier,
                            yy[:, :, None] / cell_size,
                            xx[:, :, None] / cell_size], axis=2).reshape(-1, 5)
    cfeat = np.zeros((k, 5), dtype=np.float32)
    for i, (cy, cx) in enumerate(centers):
        cfeat[i] = feats[int(cy) * w + int(cx)]
    lab = np.zeros(h * w, dtype=np.int64)
    for _ in range(n_iter):
        d = ((feats[:, None, :] - cfeat[None, :, :]) ** 2).sum(-1)
        lab = d.argmin(1)
        for i in range(k):
            sel = feats[lab == i]
            if len(sel):
                cfeat[i] = sel.mean(0)
    return lab.reshape(h, w)
