"""KernelSHAP explainers (KernelSHAPBase.scala:36; concrete classes
TabularSHAP:16, VectorSHAP, ImageSHAP, TextSHAP).

Per row: sample coalitions (exact small-|z| enumeration, default budget
2*m+2048 — KernelSHAPBase.scala:135), build perturbed samples, score ALL of
them through the model in one batched pass (GBDT models skip the DataFrame
entirely via score_matrix), MEAN-aggregate each coalition over background
replacements (the reference's per-coalition aggregation,
KernelSHAPBase.scala:69-93 — leftover sample budget becomes background
draws, cycling the full background exactly when it fits), then solve the
constrained weighted least squares for every row and class at once (the
coalition design is shared, so one factorization serves the whole chunk).
Output per row: (n_classes, m+1) — [base_value, phi_1..phi_m] per class."""
from __future__ import annotations


import numpy as np
import pandas as pd
import torch

from ..core.param import Param, toInt, toList
from ..core.registry import register
from ..core.schema import matrix_to_vector_column, vector_column_to_matrix
from .base import LocalExplainer
from .regression import (batched_kernel_shap_solve,
                         constrained_kernel_shap_solve)
from .sampler import (ImageSampler, TextSampler, VectorSampler,
                      sample_coalitions, slic_superpixels)


def _mean_over_draws(scores, n_rows, nZ, B):
    """Collapse B background draws per coalition to their mean: the head
    (n_rows*nZ*B) perturbation scores reduce to n_rows*nZ coalition values;
    the tail (background + full rows) passes through untouched."""
    if B == 1:
        return scores
    head = scores[: n_rows * nZ * B]
    tail = scores[n_rows * nZ * B:]
    head = head.reshape(n_rows * nZ, B, -1).mean(axis=1)
    return np.concatenate([head, tail])


class KernelSHAPBase(LocalExplainer):
    def _default_samples(self, m):
        return 2 * m + 2048

    def _solve_batch(self, Z, scores, nZ, n_pert, n_bg, w, v_null,
                     n_rows):
        """All rows of a chunk share Z/w: one factorization, every
        row × class solved in a single matrix solve."""
        C = scores.shape[1]
        Zt = torch.from_numpy(Z.astype(np.float64))
        wt = torch.from_numpy(w)
        per_class = []
        for k in range(C):
            V = torch.from_numpy(
                scores[:n_pert, k].reshape(n_rows, nZ).astype(np.float64))
            v_fulls = torch.from_numpy(
                scores[n_pert + n_bg:, k].astype(np.float64))
            per_class.append(batched_kernel_shap_solve(
                Zt, V, wt, float(v_null[k]), v_fulls).numpy())
        return [np.stack([np.concatenate([[float(v_null[k])],
                                          per_class[k][i]])
                          for k in range(C)])
                for i in range(n_rows)]

    def _solve(self, Z, v, w, v_null, v_full):
        phis = []
        for k in range(v.shape[1]):
            phi = constrained_kernel_shap_solve(
                torch.from_numpy(Z.astype(np.float64)),
                torch.from_numpy(v[:, k].astype(np.float64)),
                torch.from_numpy(w), float(v_null[k]), float(v_full[k]))
            phis.append(np.concatenate([[float(v_null[k])],
                                        phi.numpy().astype(np.float64)]))
        return np.stack(phis)  # (n_classes, m+1)


@register
class TabularSHAP(KernelSHAPBase):
    inputCols = Param("inputCols", "feature columns to explain", None, toList)
    backgroundData = Param("backgroundData", "background DataFrame", None,
                           is_complex=True)

    def _transform(self, df: pd.DataFrame) -> pd.DataFrame:
        cols = self.get("inputCols")
        m = len(cols)
        bg_df = self.get("backgroundData")
        bg = bg_df[cols].to_numpy(dtype=np.float64)
        rng = np.random.default_rng(self.get("seed"))
        n_samp = self.get("numSamples") or self._default_samples(m)
        Z, w = sample_coalitions(m, n_samp, rng)
        nZ = Z.shape[0]
        # reference semantics: v(S) is the MEAN model output over background
        # replacements (KernelSHAPBase.scala:69-93 aggregates per coalition);
        # spend the leftover sample budget on background draws per coalition
        B = max(1, min(len(bg), n_samp // max(1, nZ)))
        Zrep = np.repeat(Z, B, axis=0)
        explanations = []
        batch = self.get("rowBatch")
        rows = df[cols].to_numpy(dtype=np.float64)
        for s in range(0, len(rows), batch):
            chunk = rows[s:s + batch]
            frames = []
            for x in chunk:
                sampler = VectorSampler(bg, rng)
                pert = sampler.apply(x, Zrep, draws_per_state=B)
                frames.append(pert)
            all_pert = np.concatenate(frames) if frames else np.zeros((0, m))
            # v_null from background mean prediction, v_full from the row
            raw_scores = self._score_matrix(
                np.concatenate([all_pert, bg, chunk]), cols)
            scores = _mean_over_draws(raw_scores, len(chunk), nZ, B)
            v_null = scores[len(chunk) * nZ:
                            len(chunk) * nZ + len(bg)].mean(axis=0)
            explanations.extend(self._solve_batch(
                Z, scores, nZ, len(chunk) * nZ, len(bg), w, v_null,
                len(chunk)))
        out = df.copy()
        out[self.get("outputCol")] = explanations
        return out


@register
class VectorSHAP(KernelSHAPBase):
    featuresCol = Param("featuresCol", "dense vector column", "features")
    backgroundData = Param("backgroundData", "background DataFrame", None,
                           is_complex=True)

    def _transform(self, df: pd.DataFrame) -> pd.DataFrame:
        fcol = self.get("featuresCol")
        bg_df = self.get("backgroundData")
        bg = vector_column_to_matrix(bg_df, fcol).astype(np.float64)
        m = bg.shape[1]
        rng = np.random.default_rng(self.get("seed"))
        n_samp = self.get("numSamples") or self._default_samples(m)
        Z, w = sample_coalitions(m, n_samp, rng)
        nZ = Z.shape[0]
        B = max(1, min(len(bg), n_samp // max(1, nZ)))
        Zrep = np.repeat(Z, B, axis=0)
        rows = vector_column_to_matrix(df, fcol).astype(np.float64)
        explanations = []
        batch = self.get("rowBatch")
        for s in range(0, len(rows), batch):
            chunk = rows[s:s + batch]
            pert_frames = [VectorSampler(bg, rng).apply(x, Zrep,
                                                        draws_per_state=B)
                           for x in chunk]
            all_pert = np.concatenate(pert_frames)
            full = np.concatenate([bg, chunk])
            raw_scores = self._score_matrix(
                np.concatenate([all_pert, full]).astype(np.float32))
            scores = _mean_over_draws(raw_scores, len(chunk), nZ, B)
            v_null = scores[len(chunk) * nZ:
                            len(chunk) * nZ + len(bg)].mean(axis=0)
            explanations.extend(self._solve_batch(
                Z, scores, nZ, len(chunk) * nZ, len(bg), w, v_null,
                len(chunk)))
        out = df.copy()
        out[self.get("outputCol")] = explanations
        return out


@register
class TextSHAP(KernelSHAPBase):
    inputCol = Param("inputCol", "text column", "text")
    tokensCol = Param("tokensCol", "output tokens column", "tokens")

    def _transform(self, df: pd.DataFrame) -> pd.DataFrame:
        rng = np.random.default_rng(self.get("seed"))
        explanations, tokens_col = [], []
        for _, row in df.iterrows():
            tokens = str(row[self.get("inputCol")]).split()
            tokens_col.append(tokens)
            m = max(len(tokens), 1)
            n_samp = self.get("numSamples") or self._default_samples(m)
            Z, w = sample_coalitions(m, n_samp, rng)
            sampler = TextSampler(tokens)
            texts = sampler.apply(Z)
            samples_df = pd.DataFrame({
                self.get("inputCol"): texts + ["", " ".join(tokens)]})
            scores = self._score_samples(samples_df)
            v = scores[:len(texts)]
            v_null = scores[len(texts)]
            v_full = scores[len(texts) + 1]
            explanations.append(self._solve(Z, v, w, v_null, v_full))
        out = df.copy()
        out[self.get("outputCol")] = explanations
        out[self.get("tokensCol")] = tokens_col
        return out


@register
class ImageSHAP(KernelSHAPBase):
    inputCol = Param("inputCol", "image column", "image")
    cellSize = Param("cellSize", "superpixel cell size", 16, toInt)
    modifier = Param("modifier", "superpixel color weight", 10.0)
    superpixelCol = Param("superpixelCol", "output segment map column",
                          "superpixels")

    def _transform(self, df: pd.DataFrame) -> pd.DataFrame:
        rng = np.random.default_rng(self.get("seed"))
        explanations, segs_col = [], []
        for _, row in df.iterrows():
            img = np.asarray(row[self.get("inputCol")])
            segments = slic_superpixels(img, self.get("cellSize"),
                                        float(self.get("modifier")))
            segs_col.append(segments)
            m = int(segments.max()) + 1
            n_samp = self.get("numSamples") or min(self._default_samples(m), 2048)
            Z, w = sample_coalitions(m, n_samp, rng)
            sampler = ImageSampler(img, segments)
            imgs = sampler.apply(Z)
            blank = np.zeros_like(img)
            samples_df = pd.DataFrame({
                self.get("inputCol"): imgs + [blank, img]})
            scores = self._score_samples(samples_df)
            v = scores[:len(imgs)]
            v_null = scores[len(imgs)]
            v_full = scores[len(imgs) + 1]
            explanations.append(self._solve(Z, v, w, v_null, v_full))
        out = df.copy()
        out[self.get("outputCol")] = explanations
        out[self.get("superpixelCol")] = segs_col
        return out
