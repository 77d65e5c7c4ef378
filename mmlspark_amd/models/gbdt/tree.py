"""Decision-tree structure: flat node arrays + treeSHAP contributions.

The trained artifact of the GBDT trainer. Node arrays (numpy, host-side
master copy; flattened torch tensors are device caches built by Booster):
  feature[i]   int32  split feature, -1 for leaves
  threshold[i] f32    raw-value threshold (go left iff x <= threshold or NaN)
  thr_bin[i]   int32  binned threshold (training-time splits)
  left/right   int32  child node index within this tree
  value[f]     f32    leaf output (internal nodes: weighted node output, for SHAP)
  count[f]     f32    global row count reaching the node (SHAP cover weights)
  gain[f]      f32    split gain (feature importance)
  leaf_index   int32  leaf ordinal, -1 for internal
"""
from __future__ import annotations

from typing import Dict, List

import numpy as np


class Tree:
    def __init__(self, feature, threshold, thr_bin, left, right, value, count,
                 gain, leaf_index, shrinkage: float = 1.0,
                 cat_offset=None, cat_words=None):
        self.feature = np.asarray(feature, dtype=np.int32)
        self.threshold = np.asarray(threshold, dtype=np.float32)
        self.thr_bin = np.asarray(thr_bin, dtype=np.int32)
        self.left = np.asarray(left, dtype=np.int32)
        self.right = np.asarray(right, dtype=np.int32)
        self.value = np.asarray(value, dtype=np.float32)
        self.count = np.asarray(count, dtype=np.float32)
        self.gain = np.asarray(gain, dtype=np.float32)
        self.leaf_index = np.asarray(leaf_index, dtype=np.int32)
        self.shrinkage = float(shrinkage)
        # categorical splits: cat_offset[node] >= 0 indexes an 8-word (256-bit)
        # category bitset in cat_words; bit b set → category b goes LEFT
        self.cat_offset = (np.full(len(self.feature), -1, dtype=np.int32)
                           if cat_offset is None
                           else np.asarray(cat_offset, dtype=np.int32))
        self.cat_words = (np.zeros(0, dtype=np.uint32) if cat_words is None
                          else np.asarray(cat_words, dtype=np.uint32))

    def is_categorical_node(self, i: int) -> bool:
        return self.cat_offset[i] >= 0

    def cat_goes_left(self, i: int, category: float) -> bool:
        b = int(category)
        if not (0 <= b < 256) or np.isnan(category):
            return True  # missing/out-of-range → left
        w = self.cat_words[self.cat_offset[i] * 8 + (b >> 5)]
        return bool((int(w) >> (b & 31)) & 1)

    @property
    def n_nodes(self) -> int:
        return len(self.feature)

    @property
    def n_leaves(self) -> int:
        return int((self.feature < 0).sum())

    def to_dict(self) -> Dict:
        return {
            "feature": self.feature.tolist(),
            "threshold": [float(x) for x in self.threshold],
            "thr_bin": self.thr_bin.tolist(),
            "left": self.left.tolist(),
            "right": self.right.tolist(),
            "value": [float(x) for x in self.value],
            "count": [float(x) for x in self.count],
            "gain": [float(x) for x in self.gain],
            "leaf_index": self.leaf_index.tolist(),
            "shrinkage": self.shrinkage,
            "cat_offset": self.cat_offset.tolist(),
            "cat_words": [int(w) for w in self.cat_words],
        }

    @staticmethod
    def from_dict(d: Dict) -> "Tree":
        return Tree(d["feature"], d["threshold"], d["thr_bin"], d["left"],
                    d["right"], d["value"], d["count"], d["gain"],
                    d["leaf_index"], d.get("shrinkage", 1.0),
                    d.get("cat_offset"), d.get("cat_words"))

    # ----------------------------------------------------------------- treeSHAP
    def shap_values(self, X: np.ndarray, scale: float = 1.0) -> np.ndarray:
        """Path-dependent TreeSHAP (Lundberg et al.) per row.

        Output (n, n_features + 1): per-feature contributions + expected
        value in the last slot — same contract as the reference's
        ``featuresShap`` column (LightGBMBooster.featuresShap,
        booster/LightGBMBooster.scala:414; output shape
        BoosterHandler.shapOutputShape :103).
        """
        n, nf = X.shape
        out = np.zeros((n, nf + 1), dtype=np.float64)
        cover = self.count
        vals = self.value * self.shrinkage * scale

        # expected value of the tree = cover-weighted mean of leaves
        def node_expected(i):
            if self.feature[i] < 0:
                return vals[i]
            cl = max(cover[self.left[i]], 1e-12)
            cr = max(cover[self.right[i]], 1e-12)
            return (node_expected(self.left[i]) * cl
                    + node_expected(self.right[i]) * cr) / (cl + cr)

        expected = node_expected(0)
        out[:, -1] = expected

        feature = self.feature
        threshold = self.threshold
        left = self.left
        right = self.right

        class Path:
            __slots__ = ("d", "z", "o", "w")

            def __init__(self, cap):
                self.d = np.empty(cap, dtype=np.int64)
                self.z = np.empty(cap, dtype=np.float64)
                self.o = np.empty(cap, dtype=np.float64)
                self.w = np.empty(cap, dtype=np.float64)

        def extend(p, length, pz, po, pi):
            p.d[length] = pi
            p.z[length] = pz
            p.o[length] = po
            p.w[length] = 1.0 if length == 0 else 0.0
            for i in range(length - 1, -1, -1):
                p.w[i + 1] += po * p.w[i] * (i + 1) / (length + 1)
                p.w[i] = pz * p.w[i] * (length - i) / (length + 1)

        def unwind(p, length, i):
            one = p.o[i]
            zero = p.z[i]
            n_ = p.w[length]
            for j in range(length - 1, -1, -1):
                if one != 0:
                    t = p.w[j]
                    p.w[j] = n_ * (length + 1) / ((j + 1) * one)
                    n_ = t - p.w[j] * zero * (length - j) / (length + 1)
                else:
                    p.w[j] = p.w[j] * (length + 1) / (zero * (length - j))
            for j in range(i, length):
                p.d[j] = p.d[j + 1]
                p.z[j] = p.z[j + 1]
                p.o[j] = p.o[j + 1]

        def unwound_sum(p, length, i):
            one = p.o[i]
            zero = p.z[i]
            total = 0.0
            n_ = p.w[length]
            for j in range(length - 1, -1, -1):
                if one != 0:
                    t = n_ * (length + 1) / ((j + 1) * one)
                    total += t
                    n_ = p.w[j] - t * zero * (length - j) / (length + 1)
                else:
                    total += p.w[j] * (length + 1) / (zero * (length - j))
            return total

        max_depth = 64

        def recurse(row, phi, node, p, length, pz, po, pi):
            p2 = Path(max_depth)
            p2.d[:length] = p.d[:length]
            p2.z[:length] = p.z[:length]
            p2.o[:length] = p.o[:length]
            p2.w[:length] = p.w[:length]
            extend(p2, length, pz, po, pi)
            length += 1
            if feature[node] < 0:
                for i in range(1, length):
                    w = unwound_sum(p2, length - 1, i)
                    phi[p2.d[i]] += w * (p2.o[i] - p2.z[i]) * vals[node]
                return
            f = feature[node]
            x = row[f]
            if self.cat_offset[node] >= 0:
                goes_left = self.cat_goes_left(node, x)
            else:
                goes_left = x <= threshold[node] or np.isnan(x)
            hot, cold = (left[node], right[node]) if goes_left \
                else (right[node], left[node])
            iz, io = 1.0, 1.0
            k = -1
            for i in range(1, length):
                if p2.d[i] == f:
                    k = i
                    break
            if k >= 0:
                iz, io = p2.z[k], p2.o[k]
                unwind(p2, length - 1, k)
                length -= 1
            cnode = max(cover[node], 1e-12)
            recurse(row, phi, hot, p2, length, iz * cover[hot] / cnode, io, f)
            recurse(row, phi, cold, p2, length, iz * cover[cold] / cnode, 0.0, f)

        for r in range(n):
            phi = np.zeros(nf, dtype=np.float64)
            recurse(X[r], phi, 0, Path(max_depth), 0, 1.0, 1.0, -1)
            out[r, :nf] += phi
        return out


def flatten_trees(trees: List[Tree]):
    """Concatenate node arrays + per-tree offsets for the device predict kernel."""
    if not trees:
        z = np.zeros(0, dtype=np.int32)
        return dict(feature=z, threshold=z.astype(np.float32), left=z, right=z,
                    value=z.astype(np.float32), leaf_index=z,
                    offsets=np.zeros(1, dtype=np.int64),
                    weights=np.zeros(0, dtype=np.float32),
                    cat_offset=z, cat_words=np.zeros(0, dtype=np.int32))
    offsets = np.zeros(len(trees) + 1, dtype=np.int64)
    for i, t in enumerate(trees):
        offsets[i + 1] = offsets[i] + t.n_nodes
    cat = lambda attr, dt: np.concatenate([getattr(t, attr).astype(dt) for t in trees])
    value = np.concatenate([(t.value * t.shrinkage).astype(np.float32) for t in trees])
    # concatenate categorical bitsets, rebasing per-tree offsets
    cat_off_parts = []
    cat_word_parts = []
    word_base = 0
    for t in trees:
        off = t.cat_offset.astype(np.int64).copy()
        off[off >= 0] += word_base // 8
        cat_off_parts.append(off.astype(np.int32))
        cat_word_parts.append(t.cat_words.astype(np.uint32))
        word_base += len(t.cat_words)
    return dict(
        feature=cat("feature", np.int32),
        threshold=cat("threshold", np.float32),
        left=cat("left", np.int32),
        right=cat("right", np.int32),
        value=value,
        leaf_index=cat("leaf_index", np.int32),
        offsets=offsets,
        weights=np.ones(len(trees), dtype=np.float32),
        cat_offset=np.concatenate(cat_off_parts),
        cat_words=np.concatenate(cat_word_parts).view(np.int32)
        if word_base else np.zeros(0, dtype=np.int32),
    )
