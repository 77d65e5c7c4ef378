"""Booster — the trained GBDT ensemble (predict / save / load / merge / SHAP).

Re-provides the capability surface of the reference's LightGBMBooster wrapper
(booster/LightGBMBooster.scala: score:390, predictLeaf:403, featuresShap:414,
saveToString:269, mergeBooster:252, getFeatureImportances:491) natively:
trees live as flat arrays, batch scoring runs through the HIP forest-traversal
kernel on GPU (ops/hip) or the torch reference on CPU — never row-at-a-time.
"""
from __future__ import annotations

import json
from typing import List, Optional

import numpy as np
import torch

from ...ops import backend
from .binning import BinMapper
from .tree import Tree, flatten_trees


class Booster:
    def __init__(self, trees: List[Tree], objective: str, n_outputs: int,
                 base_score: np.ndarray, n_features: int,
                 feature_names: Optional[List[str]] = None,
                 bin_mapper: Optional[BinMapper] = None,
                 sigmoid: float = 1.0,
                 tree_weights: Optional[np.ndarray] = None):
        self.trees = trees
        self.objective = objective
        self.n_outputs = n_outputs
        self.base_score = np.asarray(base_score, dtype=np.float32).reshape(-1)
        self.n_features = n_features
        self.feature_names = feature_names or [f"f{i}" for i in range(n_features)]
        self.bin_mapper = bin_mapper
        self.sigmoid = sigmoid
        self.tree_weights = (np.ones(len(trees), dtype=np.float32)
                             if tree_weights is None else
                             np.asarray(tree_weights, dtype=np.float32))
        self._flat_cache = {}

    # ------------------------------------------------------------------ predict
    def _flat(self, device):
        key = str(device)
        if key not in self._flat_cache:
            f = flatten_trees(self.trees)
            self._flat_cache[key] = {
                k: torch.from_numpy(np.ascontiguousarray(v)).to(device)
                for k, v in f.items()}
            self._flat_cache[key]["weights"] = torch.from_numpy(
                self.tree_weights.copy()).to(device)
        return self._flat_cache[key]

    def invalidate_cache(self):
        self._flat_cache = {}

    @property
    def num_trees(self) -> int:
        return len(self.trees)

    @property
    def num_iterations(self) -> int:
        return len(self.trees) // max(1, self.n_outputs)

    best_iteration: int = -1  # set by early stopping; used at predict

    def predict_raw(self, X: torch.Tensor, start_iteration: int = 0,
                    num_iteration: int = -1) -> torch.Tensor:
        """Raw margin scores (n, n_outputs). With early stopping, scoring
        uses trees up to best_iteration (LightGBM semantics) unless
        num_iteration overrides."""
        if num_iteration < 0 and getattr(self, "best_iteration", -1) >= 0:
            num_iteration = self.best_iteration + 1
        if hasattr(X, "densify_chunks"):  # CsrMatrix: bounded-memory scoring
            outs = [self.predict_raw(chunk, start_iteration, num_iteration)
                    for _, chunk in X.densify_chunks()]
            if not outs:
                base = torch.from_numpy(self.base_score).to(X.device)
                return base.expand(0, self.n_outputs).clone()
            return torch.cat(outs, dim=0)
        X = X if isinstance(X, torch.Tensor) else torch.as_tensor(X, dtype=torch.float32)
        X = X.float()
        if not self.trees:
            base = torch.from_numpy(self.base_score).to(X.device)
            return base.expand(X.shape[0], self.n_outputs).clone()
        f = self._flat(X.device)
        raw = backend.predict_forest(
            f["feature"], f["threshold"], f["left"], f["right"], f["value"],
            f["offsets"], X.contiguous(), self.n_outputs, f["weights"],
            start_tree=start_iteration * self.n_outputs,
            num_iteration=num_iteration,
            cat_offset=f.get("cat_offset"), cat_words=f.get("cat_words"))
        return raw + torch.from_numpy(self.base_score).to(X.device)

    def predict_prob(self, X: torch.Tensor, start_iteration: int = 0,
                     num_iteration: int = -1) -> torch.Tensor:
        raw = self.predict_raw(X, start_iteration, num_iteration)
        if self.objective == "binary":
            p1 = torch.sigmoid(self.sigmoid * raw)
            return torch.cat([1 - p1, p1], dim=-1)
        if self.objective in ("multiclass", "softmax"):
            return torch.softmax(raw, dim=-1)
        return raw

    def predict_leaf(self, X: torch.Tensor) -> torch.Tensor:
        if hasattr(X, "densify_chunks"):  # CsrMatrix
            return torch.cat([self.predict_leaf(chunk)
                              for _, chunk in X.densify_chunks()], dim=0)
        X = X.float()
        f = self._flat(X.device)
        return backend.predict_leaf(f["feature"], f["threshold"], f["left"],
                                    f["right"], f["leaf_index"], f["offsets"],
                                    X.contiguous(),
                                    cat_offset=f.get("cat_offset"),
                                    cat_words=f.get("cat_words"))

    def _tree_depth(self) -> int:
        best = 0
        for t in self.trees:
            depth = np.zeros(t.n_nodes, dtype=np.int32)
            for i in range(t.n_nodes):  # parents precede children
                if t.feature[i] >= 0:
                    depth[t.left[i]] = depth[i] + 1
                    depth[t.right[i]] = depth[i] + 1
            if t.n_nodes:
                best = max(best, int(depth.max()))
        return best

    def expected_value(self) -> np.ndarray:
        """Cover-weighted mean output per class (+ base score), shape (K,)."""
        totals = self.base_score.astype(np.float64).copy() \
            if self.base_score.size else np.zeros(self.n_outputs)
        if totals.size < self.n_outputs:
            totals = np.resize(totals, self.n_outputs)
        for ti, (t, w) in enumerate(zip(self.trees, self.tree_weights)):
            vals = t.value * t.shrinkage * float(w)
            cover = t.count

            def rec(i):
                if t.feature[i] < 0:
                    return vals[i]
                cl = max(cover[t.left[i]], 1e-12)
                cr = max(cover[t.right[i]], 1e-12)
                return (rec(t.left[i]) * cl + rec(t.right[i]) * cr) / (cl + cr)

            totals[ti % self.n_outputs] += rec(0) if t.n_nodes else 0.0
        return totals

    def predict_contrib(self, X: torch.Tensor) -> np.ndarray:
        """TreeSHAP contributions with the expected value last per class:
        (n, n_features+1) for single-output models, (n, K*(n_features+1))
        for multiclass (LightGBM contrib layout).
        GPU path: tree_shap_k kernel (trees deeper than 32 fall back to CPU)."""
        depth = self._tree_depth() if self.trees else 0
        K = self.n_outputs
        nf = self.n_features
        expected = self.expected_value()
        if isinstance(X, torch.Tensor) and X.is_cuda and self.trees \
                and depth < 32:
            from ...ops.backend import _require_ext
            f = self._flat(X.device)
            # fold per-tree weights into leaf values for the kernel
            fw = flatten_trees(self.trees)
            scaled = fw["value"].copy()
            for t in range(len(self.trees)):
                s, e = fw["offsets"][t], fw["offsets"][t + 1]
                scaled[s:e] *= float(self.tree_weights[t])
            val = torch.from_numpy(scaled).to(X.device)
            cnt = torch.from_numpy(
                np.concatenate([t.count for t in self.trees])
                .astype(np.float32)).to(X.device)
            Xc = X.float().contiguous()
            outs = []
            row_batch = 8192  # bound per-launch runtime
            for s0 in range(0, Xc.shape[0], row_batch):
                outs.append(_require_ext().tree_shap(
                    f["feature"], f["threshold"], f["left"], f["right"], val,
                    cnt, f["offsets"], Xc[s0:s0 + row_batch], K, depth,
                    f.get("cat_offset"), f.get("cat_words")))
            res = torch.cat(outs).cpu().numpy().astype(np.float64)
            res = res.reshape(-1, K, nf + 1)
            res[:, :, -1] = expected[None, :]
            return res.reshape(-1, K * (nf + 1)).astype(np.float32) if K > 1 \
                else res[:, 0].astype(np.float32)
        Xn = X.cpu().numpy() if isinstance(X, torch.Tensor) else np.asarray(X)
        Xn = Xn.astype(np.float32)
        out = np.zeros((Xn.shape[0], K, nf + 1), dtype=np.float64)
        for ti, (t, w) in enumerate(zip(self.trees, self.tree_weights)):
            out[:, ti % K] += t.shap_values(Xn, scale=float(w))
        out[:, :, -1] = expected[None, :]
        return out.reshape(-1, K * (nf + 1)).astype(np.float32) if K > 1 \
            else out[:, 0].astype(np.float32)

    # ------------------------------------------------------------- importances
    def feature_importances(self, importance_type: str = "split") -> np.ndarray:
        """split = number of uses; gain = total gain (reference
        getFeatureImportances, booster/LightGBMBooster.scala:491)."""
        out = np.zeros(self.n_features, dtype=np.float64)
        for t in self.trees:
            mask = t.feature >= 0
            if importance_type == "gain":
                np.add.at(out, t.feature[mask], t.gain[mask])
            else:
                np.add.at(out, t.feature[mask], 1.0)
        return out

    # ------------------------------------------------------------- persistence
    def to_dict(self) -> dict:
        return {
            "version": 1,
            "objective": self.objective,
            "n_outputs": self.n_outputs,
            "base_score": [float(x) for x in self.base_score],
            "n_features": self.n_features,
            "feature_names": self.feature_names,
            "sigmoid": self.sigmoid,
            "tree_weights": [float(x) for x in self.tree_weights],
            "best_iteration": int(getattr(self, "best_iteration", -1)),
            "trees": [t.to_dict() for t in self.trees],
            "bin_mapper": (None if self.bin_mapper is None else {
                "upper_bounds": self.bin_mapper.upper_bounds.cpu().numpy().tolist(),
                "n_bins": self.bin_mapper.n_bins,
                "categorical": list(self.bin_mapper.categorical)}),
        }

    def save_to_string(self) -> str:
        """Native-model text export (analog saveToString / saveNativeModel,
        LightGBMClassifier.scala:185-205)."""
        return json.dumps(self.to_dict())

    @staticmethod
    def from_dict(d: dict) -> "Booster":
        bm = None
        if d.get("bin_mapper"):
            ub = torch.tensor(d["bin_mapper"]["upper_bounds"], dtype=torch.float32)
            bm = BinMapper(ub, int(d["bin_mapper"]["n_bins"]),
                           d["bin_mapper"].get("categorical"))
        b = Booster(
            trees=[Tree.from_dict(t) for t in d["trees"]],
            objective=d["objective"], n_outputs=int(d["n_outputs"]),
            base_score=np.array(d["base_score"], dtype=np.float32),
            n_features=int(d["n_features"]),
            feature_names=d.get("feature_names"),
            bin_mapper=bm, sigmoid=d.get("sigmoid", 1.0),
            tree_weights=np.array(d.get("tree_weights", []), dtype=np.float32)
            if d.get("tree_weights") else None)
        b.best_iteration = int(d.get("best_iteration", -1))
        return b

    @staticmethod
    def load_from_string(s: str) -> "Booster":
        """Accepts either this framework's JSON model or stock LightGBM
        native model text (``tree\\nversion=v3...``) — the reference's
        setModelString takes LightGBM text (LightGBMClassifier.scala:150),
        so models trained elsewhere score here unchanged."""
        if s.lstrip().startswith("{"):
            return Booster.from_dict(json.loads(s))
        return _from_lightgbm_text(s)

    def merge(self, other: "Booster") -> "Booster":
        """Append another booster's trees (mergeBooster analog,
        booster/LightGBMBooster.scala:252 — used by numBatches training)."""
        assert other.n_outputs == self.n_outputs
        self.trees = self.trees + other.trees
        self.tree_weights = np.concatenate([self.tree_weights, other.tree_weights])
        self.invalidate_cache()
        return self


def _to_lightgbm_text(booster: "Booster") -> str:
    """Export in LightGBM's native text model format (v3) so models trained
    here load in stock LightGBM tooling — the interop the reference's
    saveNativeModel gives its users (LightGBMClassifier.scala:185-205).
    Categorical bitset splits export with decision_type=1 (==) per LightGBM's
    categorical encoding via the cat_boundaries/cat_threshold arrays."""
    sig = float(getattr(booster, "sigmoid", 1.0))
    obj = {"binary": f"binary sigmoid:{sig:g}", "multiclass": "multiclass",
           "regression": "regression"}.get(booster.objective,
                                           booster.objective)
    lines = ["tree", "version=v3",
             f"num_class={booster.n_outputs}",
             "num_tree_per_iteration=%d" % booster.n_outputs,
             "label_index=0",
             f"max_feature_idx={booster.n_features - 1}",
             f"objective={obj}",
             "feature_names=" + " ".join(booster.feature_names),
             "feature_infos=" + " ".join(["none"] * booster.n_features), ""]
    for ti, (tree, w) in enumerate(zip(booster.trees, booster.tree_weights)):
        internal = np.nonzero(tree.feature >= 0)[0]
        leaves = np.nonzero(tree.feature < 0)[0]
        n_int = len(internal)
        imap = {int(n): i for i, n in enumerate(internal)}
        lmap = {int(n): int(tree.leaf_index[n]) for n in leaves}

        def child_ref(n):
            n = int(n)
            return imap[n] if tree.feature[n] >= 0 else ~lmap[n]

        split_feature, threshold, decision_type = [], [], []
        left_child, right_child = [], []
        internal_value, internal_count = [], []
        cat_boundaries = [0]
        cat_threshold = []
        for n in internal:
            split_feature.append(int(tree.feature[n]))
            if tree.cat_offset[n] >= 0:
                decision_type.append(1)  # categorical ==
                threshold.append(float(len(cat_boundaries) - 1))
                words = tree.cat_words[tree.cat_offset[n] * 8:
                                       tree.cat_offset[n] * 8 + 8]
                cat_threshold.extend(int(x) for x in words)
                cat_boundaries.append(len(cat_threshold))
            else:
                decision_type.append(2)  # numerical <= with default-left
                threshold.append(float(tree.threshold[n]))
            left_child.append(child_ref(tree.left[n]))
            right_child.append(child_ref(tree.right[n]))
            internal_value.append(float(tree.value[n]))
            internal_count.append(int(tree.count[n]))
        leaf_sorted = sorted(leaves, key=lambda n: tree.leaf_index[n])
        # stock LightGBM has no base-score field: it bakes the init score
        # into the first iteration's leaves (GBDT::AddBias on tree 0) — do
        # the same so text predictions match predict_raw everywhere
        bias = (float(booster.base_score[ti % booster.n_outputs])
                if ti < booster.n_outputs else 0.0)
        leaf_value = [float(tree.value[n] * tree.shrinkage * w) + bias
                      for n in leaf_sorted]
        leaf_count = [int(tree.count[n]) for n in leaf_sorted]
        lines += [f"Tree={ti}",
                  f"num_leaves={len(leaves)}",
                  "num_cat=%d" % (len(cat_boundaries) - 1),
                  "split_feature=" + " ".join(map(str, split_feature)),
                  "threshold=" + " ".join(f"{t:.17g}" for t in threshold),
                  "decision_type=" + " ".join(map(str, decision_type)),
                  "left_child=" + " ".join(map(str, left_child)),
                  "right_child=" + " ".join(map(str, right_child)),
                  "leaf_value=" + " ".join(f"{v:.17g}" for v in leaf_value),
                  "leaf_count=" + " ".join(map(str, leaf_count)),
                  "internal_value=" + " ".join(f"{v:.17g}"
                                               for v in internal_value),
                  "internal_count=" + " ".join(map(str, internal_count)),
                  "shrinkage=%g" % booster.trees[ti].shrinkage]
        if cat_threshold:
            lines.insert(len(lines) - 1, "cat_boundaries=" +
                         " ".join(map(str, cat_boundaries)))
            lines.insert(len(lines) - 1, "cat_threshold=" +
                         " ".join(map(str, cat_threshold)))
        lines.append("")
    lines.append("end of trees")
    lines.append("")
    return "\n".join(lines)


def _from_lightgbm_text(s: str) -> "Booster":
    """Parse LightGBM native model text (v2/v3) into a Booster — the inverse
    of to_lightgbm_text, and the interop the reference's setModelString
    expects (LightGBMClassifier.scala:150: users pass stock LightGBM model
    strings for warm start / scoring).  Internal nodes keep LightGBM's
    node numbering (node i = internal i), leaves are appended after the
    internal nodes.  Categorical bitsets wider than 256 bits are rejected
    (our device bitset is 8x32 bits; max_bin here is <= 255)."""
    header = {}
    tree_blocks = []
    cur = None
    for line in s.splitlines():
        line = line.strip()
        if not line or line == "end of trees":
            continue
        if line.startswith("Tree="):
            cur = {}
            tree_blocks.append(cur)
            continue
        if "=" not in line:
            if cur is None:
                header[line] = True  # bare "tree" magic
            continue
        k, v = line.split("=", 1)
        (header if cur is None else cur)[k] = v

    n_outputs = int(header.get("num_class", 1))
    n_features = int(header["max_feature_idx"]) + 1
    obj_parts = header.get("objective", "regression").split()
    objective = obj_parts[0]
    sigmoid = 1.0
    for p in obj_parts[1:]:
        if p.startswith("sigmoid:"):
            sigmoid = float(p.split(":")[1])
    feature_names = header.get(
        "feature_names", " ".join(f"f{i}" for i in range(n_features))).split()

    def ints(block, key, default=""):
        v = block.get(key, default)
        return [int(float(x)) for x in v.split()] if v else []

    def floats(block, key, default=""):
        v = block.get(key, default)
        return [float(x) for x in v.split()] if v else []

    trees = []
    for tb in tree_blocks:
        split_feature = ints(tb, "split_feature")
        threshold = floats(tb, "threshold")
        decision_type = ints(tb, "decision_type")
        lchild = ints(tb, "left_child")
        rchild = ints(tb, "right_child")
        leaf_value = floats(tb, "leaf_value")
        leaf_count = ints(tb, "leaf_count")
        internal_value = floats(tb, "internal_value")
        internal_count = ints(tb, "internal_count")
        split_gain = floats(tb, "split_gain")
        cat_boundaries = ints(tb, "cat_boundaries")
        cat_threshold = ints(tb, "cat_threshold")
        n_int = len(split_feature)
        n_leaf = len(leaf_value)
        if n_int == 0:  # single-leaf (stump) tree
            trees.append(Tree([-1], [0.0], [0], [-1], [-1],
                              [leaf_value[0] if leaf_value else 0.0],
                              [leaf_count[0] if leaf_count else 0.0],
                              [0.0], [0]))
            continue
        n_nodes = n_int + n_leaf

        def ref(r):  # LightGBM child ref: >=0 internal, <0 → leaf ~r
            return r if r >= 0 else n_int + (~r)

        feature = np.full(n_nodes, -1, dtype=np.int32)
        thr = np.zeros(n_nodes, dtype=np.float32)
        left = np.full(n_nodes, -1, dtype=np.int32)
        right = np.full(n_nodes, -1, dtype=np.int32)
        value = np.zeros(n_nodes, dtype=np.float32)
        count = np.zeros(n_nodes, dtype=np.float32)
        gain = np.zeros(n_nodes, dtype=np.float32)
        leaf_index = np.full(n_nodes, -1, dtype=np.int32)
        cat_offset = np.full(n_nodes, -1, dtype=np.int32)
        cat_words = []
        for i in range(n_int):
            feature[i] = split_feature[i]
            left[i] = ref(lchild[i])
            right[i] = ref(rchild[i])
            if internal_value:
                value[i] = internal_value[i]
            if internal_count:
                count[i] = internal_count[i]
            if split_gain:
                gain[i] = split_gain[i]
            if decision_type and (decision_type[i] & 1):  # categorical
                ci = int(threshold[i])
                lo, hi = cat_boundaries[ci], cat_boundaries[ci + 1]
                if hi - lo > 8:
                    raise ValueError(
                        "categorical bitset wider than 256 bits unsupported")
                words = cat_threshold[lo:hi] + [0] * (8 - (hi - lo))
                cat_offset[i] = len(cat_words) // 8
                cat_words.extend(words)
            else:
                thr[i] = threshold[i]
        for j in range(n_leaf):
            value[n_int + j] = leaf_value[j]
            if leaf_count:
                count[n_int + j] = leaf_count[j]
            leaf_index[n_int + j] = j
        trees.append(Tree(feature, thr, np.zeros(n_nodes, dtype=np.int32),
                          left, right, value, count, gain, leaf_index,
                          shrinkage=1.0,  # leaf_value already includes it
                          cat_offset=cat_offset,
                          cat_words=np.array(cat_words, dtype=np.uint64)
                          .astype(np.uint32) if cat_words else None))

    return Booster(trees=trees, objective=objective, n_outputs=n_outputs,
                   base_score=np.zeros(n_outputs, dtype=np.float32),
                   n_features=n_features, feature_names=feature_names,
                   bin_mapper=None, sigmoid=sigmoid,
                   tree_weights=np.ones(len(trees), dtype=np.float32))


Booster.to_lightgbm_text = _to_lightgbm_text
