"""LIME explainers (LIMEBase.scala:49; TabularLIME:17, VectorLIME, ImageLIME,
TextLIME + the legacy Superpixel transformer of core/.../lime/).

Per row: Bernoulli on/off states → perturbed samples → batched scoring →
kernel-weighted lasso per target class (kernel weight
sqrt(exp(-(d/width)^2)), LIMEBase.scala:55-65; per-row lasso :96-113).
Output per row: (n_classes, m) coefficient matrix."""
from __future__ import annotations

import numpy as np
import pandas as pd
import torch

from ..core.param import Param, toFloat, toInt, toList
from ..core.registry import register
from ..core.schema import matrix_to_vector_column, vector_column_to_matrix
from .base import LocalExplainer
from .regression import lasso_regression
from .sampler import (ImageSampler, TextSampler, VectorSampler,
                      lime_sample_states, slic_superpixels)


class LIMEBase(LocalExplainer):
    kernelWidth = Param("kernelWidth", "RBF kernel width on state distance",
                        0.75, toFloat)
    regularization = Param("regularization", "lasso alpha", 0.0, toFloat)

    def _default_samples(self, m):
        return 1000

    def _weights(self, states: np.ndarray) -> np.ndarray:
        d = np.sqrt(((1.0 - states) ** 2).sum(axis=1)) / max(states.shape[1], 1) ** 0.5
        w = np.exp(-(d / self.get("kernelWidth")) ** 2)
        return np.sqrt(w)

    def _fit_states(self, states, scores, r2_out=None):
        """Weighted lasso per class; optionally records the weighted R² of
        the local surrogate (the reference's metricsCol output,
        LIMEBase.scala r2 column)."""
        w = self._weights(states)
        coefs = []
        r2s = []
        Xs = torch.from_numpy(states.astype(np.float64))
        wt = torch.from_numpy(w)
        for k in range(scores.shape[1]):
            yv = torch.from_numpy(scores[:, k].astype(np.float64))
            res = lasso_regression(Xs, yv, alpha=self.get("regularization"),
                                   sample_weight=wt)
            coefs.append(res.coefficients.numpy())
            if r2_out is not None:
                r2s.append(float(res.r_squared))
        if r2_out is not None:
            r2_out.append(np.asarray(r2s))
        return np.stack(coefs)  # (n_classes, m)


@register
class TabularLIME(LIMEBase):
    inputCols = Param("inputCols", "feature columns", None, toList)
    backgroundData = Param("backgroundData", "background DataFrame", None,
                           is_complex=True)

    def _transform(self, df: pd.DataFrame) -> pd.DataFrame:
        cols = self.get("inputCols")
        m = len(cols)
        bg = self.get("backgroundData")[cols].to_numpy(dtype=np.float64)
        rng = np.random.default_rng(self.get("seed"))
        n_samp = self.get("numSamples") or self._default_samples(m)
        explanations = []
        r2s = []
        rows = df[cols].to_numpy(dtype=np.float64)
        batch = self.get("rowBatch")
        for s0 in range(0, len(rows), batch):
            chunk = rows[s0:s0 + batch]
            states_l = [lime_sample_states(m, n_samp, rng) for _ in chunk]
            perts = [VectorSampler(bg, rng).apply(x, st)
                     for x, st in zip(chunk, states_l)]
            # one scoring fan-out for the whole chunk
            scores = self._score_matrix(np.concatenate(perts), cols)                 if perts else np.zeros((0, 1))
            o = 0
            for st, p_ in zip(states_l, perts):
                explanations.append(
                    self._fit_states(st, scores[o:o + len(p_)],
                                     r2_out=r2s))
                o += len(p_)
        out = df.copy()
        out[self.get("outputCol")] = explanations
        if self.get("metricsCol"):
            out[self.get("metricsCol")] = r2s
        return out


@register
class VectorLIME(LIMEBase):
    featuresCol = Param("featuresCol", "dense vector column", "features")
    backgroundData = Param("backgroundData", "background DataFrame", None,
                           is_complex=True)

    def _transform(self, df: pd.DataFrame) -> pd.DataFrame:
        fcol = self.get("featuresCol")
        bg = vector_column_to_matrix(self.get("backgroundData"), fcol)
        bg = bg.astype(np.float64)
        m = bg.shape[1]
        rng = np.random.default_rng(self.get("seed"))
        n_samp = self.get("numSamples") or self._default_samples(m)
        rows = vector_column_to_matrix(df, fcol).astype(np.float64)
        explanations = []
        r2s = []
        batch = self.get("rowBatch")
        for s0 in range(0, len(rows), batch):
            chunk = rows[s0:s0 + batch]
            states_l = [lime_sample_states(m, n_samp, rng) for _ in chunk]
            perts = [VectorSampler(bg, rng).apply(x, st).astype(np.float32)
                     for x, st in zip(chunk, states_l)]
            scores = self._score_matrix(np.concatenate(perts))                 if perts else np.zeros((0, 1))
            o = 0
            for st, p_ in zip(states_l, perts):
                explanations.append(
                    self._fit_states(st, scores[o:o + len(p_)],
                                     r2_out=r2s))
                o += len(p_)
        out = df.copy()
        out[self.get("outputCol")] = explanations
        if self.get("metricsCol"):
            out[self.get("metricsCol")] = r2s
        return out


@register
class TextLIME(LIMEBase):
    inputCol = Param("inputCol", "text column", "text")
    tokensCol = Param("tokensCol", "output tokens column", "tokens")

    def _transform(self, df: pd.DataFrame) -> pd.DataFrame:
        rng = np.random.default_rng(self.get("seed"))
        explanations, tokens_col, r2s = [], [], []
        for _, row in df.iterrows():
            tokens = str(row[self.get("inputCol")]).split()
            tokens_col.append(tokens)
            m = max(len(tokens), 1)
            n_samp = self.get("numSamples") or self._default_samples(m)
            states = lime_sample_states(m, n_samp, rng)
            texts = TextSampler(tokens).apply(states)
            scores = self._score_samples(
                pd.DataFrame({self.get("inputCol"): texts}))
            explanations.append(self._fit_states(states, scores,
                                                 r2_out=r2s))
        out = df.copy()
        out[self.get("outputCol")] = explanations
        if self.get("metricsCol"):
            out[self.get("metricsCol")] = r2s
        out[self.get("tokensCol")] = tokens_col
        return out


@register
class ImageLIME(LIMEBase):
    inputCol = Param("inputCol", "image column", "image")
    cellSize = Param("cellSize", "superpixel cell size", 16, toInt)
    modifier = Param("modifier", "superpixel color weight", 10.0, toFloat)
    superpixelCol = Param("superpixelCol", "segment map column", "superpixels")

    def _transform(self, df: pd.DataFrame) -> pd.DataFrame:
        rng = np.random.default_rng(self.get("seed"))
        explanations, segs_col, r2s = [], [], []
        for _, row in df.iterrows():
            img = np.asarray(row[self.get("inputCol")])
            segments = slic_superpixels(img, self.get("cellSize"),
                                        self.get("modifier"))
            segs_col.append(segments)
            m = int(segments.max()) + 1
            n_samp = self.get("numSamples") or min(self._default_samples(m), 512)
            states = lime_sample_states(m, n_samp, rng)
            imgs = ImageSampler(img, segments).apply(states)
            scores = self._score_samples(
                pd.DataFrame({self.get("inputCol"): imgs}))
            explanations.append(self._fit_states(states, scores,
                                                 r2_out=r2s))
        out = df.copy()
        out[self.get("outputCol")] = explanations
        if self.get("metricsCol"):
            out[self.get("metricsCol")] = r2s
        out[self.get("superpixelCol")] = segs_col
        return out


@register
class SuperpixelTransformer(LIMEBase):
    """Standalone superpixel segmentation (lime/SuperpixelTransformer.scala)."""
    inputCol = Param("inputCol", "image column", "image")
    outputCol = Param("outputCol", "segment map column", "superpixels")
    cellSize = Param("cellSize", "superpixel cell size", 16, toInt)
    modifier = Param("modifier", "superpixel color weight", 10.0, toFloat)

    def _transform(self, df: pd.DataFrame) -> pd.DataFrame:
        out = df.copy()
        out[self.get("outputCol")] = [
            slic_superpixels(np.asarray(v), self.get("cellSize"),
                             self.get("modifier"))
            for v in df[self.get("inputCol")].to_numpy()]
        return out
