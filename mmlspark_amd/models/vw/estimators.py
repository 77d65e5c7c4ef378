"""VW-equivalent estimators: hashed sparse online learning on MI355X.

Parity targets: VowpalWabbitClassifier:21 / VowpalWabbitRegressor
/ VowpalWabbitContextualBandit:106 (vw/src/main/scala/.../VowpalWabbitClassifier.scala
etc.).  The native VW example.learn loop (VowpalWabbitBase.trainRow:261-292)
becomes minibatched adaptive sparse-SGD HIP kernels over a 2^b weight table in
HBM; the end-of-pass spanning-tree AllReduce (VowpalWabbitBase.scala:363-368)
becomes ONE RCCL all_reduce of the dense weight table over xGMI (2^18 floats =
1 MB — latency-bound, single launch).  Per-partition perf timers surface as a
performance-statistics DataFrame (TrainingStats parity,
VowpalWabbitBase.scala:27-46,464-490).
"""
from __future__ import annotations

import time
from typing import Optional

import numpy as np
import pandas as pd
import torch

from ...core.param import Param, Params, toBool, toFloat, toInt, toString
from ...core.pipeline import Estimator, Model
from ...core.registry import register
from ...core.schema import SparseVector, matrix_to_vector_column
from ...ops import backend
from ...parallel.comm import get_comm
from ...utils.devices import default_device

_LOSS_BY_NAME = {"squared": "squared", "logistic": "logistic", "hinge": "hinge"}


def _extract_csr(df: pd.DataFrame, col: str, extra_cols=None, device="cpu"):
    """SparseVector / dense-vector column(s) -> CSR tensors on device.

    Columns are combined VW-style: the primary column keeps its indices
    (default namespace), every additional column's indices are offset by a
    murmur3 seed of its name (namespace seeding,
    VowpalWabbitMurmurWithPrefix semantics) so that columns of equal size
    never alias onto the same table slots after the 2^b mask."""
    from .murmur import hash_string
    cols = [col] + list(extra_cols or [])
    idx_parts, val_parts, counts = [], [], np.zeros(len(df), dtype=np.int64)
    total_dims = 0
    for ci, c in enumerate(cols):
        seed = 0 if ci == 0 else (hash_string(c) & 0x3FFFFFFF)
        vals = df[c].to_numpy()
        if len(vals) and isinstance(vals[0], SparseVector):
            total_dims += vals[0].size
        else:
            total_dims += len(np.asarray(vals[0])) if len(vals) else 0
        for i, v in enumerate(vals):
            if isinstance(v, SparseVector):
                vi = (v.indices.astype(np.int64) + seed) & 0x3FFFFFFF
                idx_parts.append((i, vi.astype(np.int32), v.values))
                counts[i] += len(v.indices)
            else:
                dense = np.asarray(v, dtype=np.float32)
                nz = np.nonzero(dense)[0]
                vi = (nz.astype(np.int64) + seed) & 0x3FFFFFFF
                idx_parts.append((i, vi.astype(np.int32), dense[nz]))
                counts[i] += len(nz)
    offset_base = total_dims
    # assemble in row order
    idx_parts.sort(key=lambda t: t[0])
    if idx_parts:
        indices = np.concatenate([p[1] for p in idx_parts])
        values = np.concatenate([p[2] for p in idx_parts])
    else:
        indices = np.zeros(0, dtype=np.int32)
        values = np.zeros(0, dtype=np.float32)
    offsets = np.zeros(len(df) + 1, dtype=np.int64)
    np.cumsum(counts, out=offsets[1:])
    return (torch.from_numpy(indices.astype(np.int32)).to(device),
            torch.from_numpy(values.astype(np.float32)).to(device),
            torch.from_numpy(offsets).to(device),
            offset_base)


def _filter_csr(idx, val, off, mask):
    """Row-filter a CSR triple (device-side, no host loop)."""
    counts = off[1:] - off[:-1]
    keep = counts[mask]
    new_off = torch.zeros(int(mask.sum()) + 1, dtype=off.dtype,
                          device=off.device)
    new_off[1:] = keep.cumsum(0)
    total = int(new_off[-1])
    if total == 0:
        return idx[:0], val[:0], new_off
    starts = off[:-1][mask]
    seg = torch.repeat_interleave(starts, keep)
    o = torch.arange(total, device=off.device)
    e = seg + (o - torch.repeat_interleave(new_off[:-1], keep))
    return idx[e], val[e], new_off


def _avg_loss(pred: torch.Tensor, y: torch.Tensor, loss: str) -> float:
    if loss == "logistic":
        ll = torch.log1p(torch.exp(-y * pred))
    elif loss == "hinge":
        ll = (1 - y * pred).clamp_min(0)
    else:  # squared
        ll = (pred - y) ** 2
    return float(ll.mean())


class _VWParams(Params):
    labelCol = Param("labelCol", "label column", "label")
    featuresCol = Param("featuresCol", "hashed features column", "features")
    additionalFeatures = Param("additionalFeatures",
                               "extra feature columns (namespaces)", None)
    weightCol = Param("weightCol", "importance weight column", None)
    predictionCol = Param("predictionCol", "prediction column", "prediction")
    numPasses = Param("numPasses", "passes over the data", 1, toInt)
    learningRate = Param("learningRate", "initial learning rate", 0.5, toFloat)
    powerT = Param("powerT", "lr decay exponent", 0.5, toFloat)
    l1 = Param("l1", "L1 regularization (end-of-pass truncation)", 0.0, toFloat)
    l2 = Param("l2", "L2 regularization", 0.0, toFloat)
    numBits = Param("numBits", "log2 weight-table size", 18, toInt)
    lossFunction = Param("lossFunction", "squared|logistic|hinge", None)
    batchSize = Param("batchSize", "SGD minibatch size (GPU hogwild window)",
                      4096, toInt)
    hashSeed = Param("hashSeed", "murmur seed (accepted for API compat; hashing "
                     "happens in VowpalWabbitFeaturizer via its own seed param — "
                     "the estimator consumes pre-hashed vectors)", 0, toInt)
    adaptive = Param("adaptive", "per-weight adaptive (AdaGrad) rates "
                     "(--adaptive)", True, toBool)
    normalized = Param("normalized", "scale updates by running max|x| per "
                       "weight (--normalized, NAG-style)", False, toBool)
    invariant = Param("invariant", "importance-weight-invariant updates "
                      "(--invariant): closed-form integration of the per-"
                      "example gradient flow, safe for large weights", False,
                      toBool)
    initialModel = Param("initialModel", "warm-start weight table", None,
                         is_complex=True)
    passThroughArgs = Param("passThroughArgs",
                            "VW-style arg string (subset parsed: --l1 --l2 "
                            "--learning_rate --power_t -b --passes "
                            "--loss_function)", "", toString)
    args = Param("args", "VW arg string (alias of passThroughArgs — the "
                 "reference's primary arg surface)", "", toString)
    testArgs = Param("testArgs", "prediction-time VW args (accepted; "
                     "scoring here has no mutable flags)", "", toString)
    interactions = Param("interactions", "feature-column groups to cross "
                         "with VW's FNV hash-combine (estimator-level -q): "
                         "each entry a list or 'colA,colB' string", None)
    ignoreNamespaces = Param("ignoreNamespaces", "first-letter namespaces "
                             "whose additionalFeatures columns are dropped",
                             None)
    useBarrierExecutionMode = Param("useBarrierExecutionMode",
                                    "gang-schedule ranks (no-op: ranks are "
                                    "always gang-launched here)", False,
                                    toBool)
    bfgs = Param("bfgs", "second-order full-batch L-BFGS optimization "
                 "(--bfgs) instead of online SGD", False, toBool)
    holdoutOff = Param("holdoutOff", "--holdout_off: train multi-pass on "
                       "every example (no holdout early termination)",
                       False, toBool)
    holdoutPeriod = Param("holdoutPeriod", "--holdout_period: every k-th "
                          "example is held out when passes>1 (VW default 10)",
                          10, toInt)
    earlyTerminate = Param("earlyTerminate", "--early_terminate: stop after "
                           "this many passes without holdout improvement",
                           3, toInt)
    maxIterBfgs = Param("maxIterBfgs", "L-BFGS iteration budget", 100, toInt)
    device = Param("device", "cpu|cuda|auto", "auto", toString)

    def _feature_frame(self, df: pd.DataFrame):
        """Apply ignoreNamespaces + estimator-level interactions; returns
        (df_with_interaction_cols, effective extra feature columns)."""
        extra = list(self.get("additionalFeatures") or [])
        ig = self.get("ignoreNamespaces") or []
        if ig:
            extra = [c for c in extra if c[:1] not in ig]
        specs = list(self.get("interactions") or [])
        specs += self._resolve_quadratic(df)  # raw-input -q crossing
        from .featurizer import VowpalWabbitInteractions
        for k, spec in enumerate(specs):
            cols = (list(spec) if isinstance(spec, (list, tuple))
                    else [c.strip() for c in str(spec).split(",")])
            out = f"__interact_{k}"
            df = VowpalWabbitInteractions(
                inputCols=cols, outputCol=out,
                numBits=self.get("numBits")).transform(df)
            extra.append(out)
        return df, extra

    def _parse_args(self):
        """Apply passThroughArgs (analog of the reference building the VW arg
        string, VowpalWabbitBase.scala:531-543 — here parsed back to Params)."""
        s = ((self.get("passThroughArgs") or "") + " "
             + (self.get("args") or "")).split()
        i = 0
        mapping = {"--l1": "l1", "--l2": "l2", "--learning_rate": "learningRate",
                   "--power_t": "powerT", "-b": "numBits", "--bit_precision":
                   "numBits", "--passes": "numPasses",
                   "--loss_function": "lossFunction",
                   "--holdout_period": "holdoutPeriod",
                   "--early_terminate": "earlyTerminate"}
        flags = {"--adaptive": "adaptive", "--normalized": "normalized",
                 "--invariant": "invariant", "--bfgs": "bfgs",
                 "--holdout_off": "holdoutOff"}
        self._quadratic_specs = []
        while i < len(s):
            if s[i] in mapping and i + 1 < len(s):
                self.set(mapping[s[i]], s[i + 1])
                i += 2
            elif s[i] in flags:
                self.set(flags[s[i]], True)
                i += 1
            elif s[i] in ("-q", "--quadratic") and i + 1 < len(s):
                # raw-input namespace crossing (VowpalWabbitBase arg
                # surface): 'ab' crosses namespaces a×b; ':' is a wildcard
                self._quadratic_specs.append(s[i + 1])
                i += 2
            else:
                i += 1

    def _resolve_quadratic(self, df: pd.DataFrame) -> list:
        """-q namespace pairs → concrete column pairs.  A namespace is a
        column's first letter (VowpalWabbitFeaturizer semantics); ':' means
        every namespace (so '-q ::' crosses all feature-column pairs)."""
        specs = getattr(self, "_quadratic_specs", None) or []
        if not specs:
            return []
        cols = [self.get("featuresCol")] + list(
            self.get("additionalFeatures") or [])
        cols = [c for c in cols if c in df.columns]
        out = []
        for spec in specs:
            a, b = (spec + "::")[:2]
            ca = cols if a == ":" else [c for c in cols if c[:1] == a]
            cb = cols if b == ":" else [c for c in cols if c[:1] == b]
            for x in ca:
                for y in cb:
                    if (x, y) not in out and (y, x) not in out:
                        out.append((x, y))
        return [list(p) for p in out]


class _VWBase(_VWParams, Estimator):
    _default_loss = "squared"
    _binary_labels = False

    def _fit_bfgs(self, idx, val, off, labels, ex_w, w0, loss, l2, comm):
        n_ex = off.numel() - 1
        counts = off[1:] - off[:-1]
        seg = torch.repeat_interleave(
            torch.arange(n_ex, device=idx.device), counts)
        il = idx.long()
        ew = ex_w if ex_w is not None else torch.ones_like(labels)
        w = w0.clone().requires_grad_(True)

        def forward():
            contrib = w[il] * val
            preds = torch.zeros(n_ex, device=w.device).index_add(
                0, seg, contrib)
            if loss == "logistic":
                per = torch.nn.functional.softplus(-labels * preds)
            elif loss == "hinge":
                per = torch.relu(1.0 - labels * preds)
            else:
                per = 0.5 * (preds - labels) ** 2
            out = (per * ew).sum() / ew.sum()
            if l2 > 0:
                out = out + l2 * (w * w).sum()
            return out

        opt = torch.optim.LBFGS([w], max_iter=self.get("maxIterBfgs"),
                                history_size=10,
                                line_search_fn="strong_wolfe")

        def closure():
            opt.zero_grad()
            ls = forward()
            ls.backward()
            if comm.is_distributed:
                # ranks must see the SAME loss value and gradient, or the
                # strong-Wolfe line searches diverge across ranks
                comm.all_reduce(w.grad)
                w.grad /= comm.world_size
                with torch.no_grad():
                    lsg = ls.detach().clone()
                    comm.all_reduce(lsg)
                    return lsg / comm.world_size
            return ls

        opt.step(closure)
        return w.detach()

    def _fit(self, df: pd.DataFrame):
        self._parse_args()
        # materialize -q namespace pairs into the interactions param so the
        # SAME crossing applies at fit and at model transform time
        # (idempotent: refitting must not duplicate pairs)
        q = self._resolve_quadratic(df)
        if q:
            existing = [list(p) if isinstance(p, (list, tuple)) else [p]
                        for p in (self.get("interactions") or [])]
            q = [p for p in q if p not in existing]
            if q:
                self.set("interactions", existing + q)
            self._quadratic_specs = []
        comm = get_comm()
        device = default_device(self.get("device"))
        t_ingest = time.perf_counter()
        df, extra_cols = self._feature_frame(df)
        idx, val, off, _ = _extract_csr(df, self.get("featuresCol"),
                                        extra_cols, device)
        # VW semantics: every feature index lands in the 2^b table (hash &
        # mask); the concatenated column offsets can exceed it otherwise
        idx = idx & ((1 << self.get("numBits")) - 1)
        y = df[self.get("labelCol")].to_numpy(dtype=np.float32)
        if self._binary_labels and getattr(self, "_convert_labels", True):
            y = np.where(y > 0, 1.0, -1.0).astype(np.float32)
        labels = torch.from_numpy(y).to(device)
        ex_w = None
        if self.get("weightCol"):
            ex_w = torch.from_numpy(
                df[self.get("weightCol")].to_numpy(dtype=np.float32)).to(device)
        ingest_s = time.perf_counter() - t_ingest

        bits = self.get("numBits")
        tbl = 1 << bits
        init = self.get("initialModel")
        if init is not None:
            w = torch.from_numpy(np.asarray(init["weights"],
                                            dtype=np.float32)).to(device).clone()
            g = torch.from_numpy(np.asarray(init["adaptive"],
                                            dtype=np.float32)).to(device).clone()
        else:
            w = torch.zeros(tbl, dtype=torch.float32, device=device)
            g = torch.zeros(tbl, dtype=torch.float32, device=device)
        s_tbl = (torch.zeros(tbl, dtype=torch.float32, device=device)
                 if self.get("normalized") else None)

        loss = self.get("lossFunction") or self._default_loss
        lr = self.get("learningRate")
        l1, l2 = self.get("l1"), self.get("l2")
        power_t = self.get("powerT") if self.get("adaptive") else 0.0
        bs = self.get("batchSize")
        n = len(df)
        learn_s = 0.0
        multipass_s = 0.0
        if self.get("bfgs"):
            # --bfgs: full-batch second-order optimization over the hashed
            # weight table (VowpalWabbitBase.scala arg surface).  The
            # forward is a differentiable gather + segment-sum, so torch's
            # L-BFGS (with line search) runs it entirely on device.
            t0 = time.perf_counter()
            w = self._fit_bfgs(idx, val, off, labels, ex_w, w, loss, l2,
                               comm)
            learn_s = time.perf_counter() - t0
        # multi-pass example cache replay (endPass/performRemainingPasses,
        # VowpalWabbitBase.scala:363-368): the cache is the device-resident
        # CSR itself.  VW holdout semantics: with passes>1 every k-th
        # example is held out of training and scores each pass; training
        # stops after earlyTerminate passes without holdout improvement
        # (disable with --holdout_off).
        n_passes = 0 if self.get("bfgs") else self.get("numPasses")
        use_holdout = n_passes > 1 and not self.get("holdoutOff")
        tr = (idx, val, off, labels, ex_w)
        ho = None
        if use_holdout:
            period = max(2, self.get("holdoutPeriod"))
            pos = torch.arange(n, device=off.device)
            ho_mask = (pos % period) == (period - 1)
            ti, tv, to = _filter_csr(idx, val, off, ~ho_mask)
            tr = (ti, tv, to, labels[~ho_mask],
                  ex_w[~ho_mask] if ex_w is not None else None)
            hi, hv, hoff = _filter_csr(idx, val, off, ho_mask)
            ho = (hi, hv, hoff, labels[ho_mask])
        t_idx, t_val, t_off, t_y, t_w = tr
        n_train = int(t_y.numel())
        best_holdout = float("inf")
        passes_run = 0
        no_improve = 0
        for p in range(n_passes):
            t0 = time.perf_counter()
            for s in range(0, n_train, bs):
                e = min(s + bs, n_train)
                o = t_off[s:e + 1] - t_off[s]
                sl = slice(int(t_off[s]), int(t_off[e]))
                backend.vw_sgd_minibatch(
                    t_idx[sl], t_val[sl], o, t_y[s:e], w, g, lr, l2,
                    power_t, loss,
                    t_w[s:e] if t_w is not None else None, s_tbl,
                    invariant=self.get("invariant"))
            learn_s += time.perf_counter() - t0
            passes_run += 1
            # end-of-pass sync: RCCL all_reduce of weights + accumulators
            t0 = time.perf_counter()
            if comm.is_distributed:
                comm.all_reduce(w)
                w /= comm.world_size
                comm.all_reduce(g)
                g /= comm.world_size
                if s_tbl is not None:
                    comm.all_reduce(s_tbl, op="max")
            if l1 > 0:  # proximal truncation
                w.copy_(torch.sign(w) * (w.abs() - lr * l1).clamp_min(0))
            multipass_s += time.perf_counter() - t0
            if ho is not None:
                hloss = _avg_loss(backend.vw_predict(ho[0], ho[1], ho[2], w),
                                  ho[3], loss)
                if comm.is_distributed:
                    t = torch.tensor([hloss, 1.0], device=w.device)
                    comm.all_reduce(t)
                    hloss = float(t[0] / t[1])
                if hloss < best_holdout - 1e-9:
                    best_holdout, no_improve = hloss, 0
                    best_snap = (w.clone(), g.clone())
                else:
                    no_improve += 1
                    if no_improve >= self.get("earlyTerminate"):
                        break
        if ho is not None and best_holdout < float("inf"):
            # keep the best-holdout pass's weights (early termination would
            # otherwise return a worse — possibly diverged — final pass)
            w, g = best_snap

        model = self._model_class()(weights=w.cpu().numpy(),
                                    adaptive=g.cpu().numpy())
        for pname in ("labelCol", "featuresCol", "additionalFeatures",
                      "predictionCol", "numBits", "interactions",
                      "ignoreNamespaces"):
            model.set(pname, self.get(pname))
        model.set("lossFunction", loss)
        model._stats = pd.DataFrame([{
            "partitionId": comm.rank,
            "ipassCurrent": (passes_run if n_passes else
                             self.get("numPasses")),
            "numberOfExamplesPerPass": n,
            "totalNumberOfFeatures": int(off[-1]),
            "nativeIngestTimeNs": int(ingest_s * 1e9),
            "learnTimeNs": int(learn_s * 1e9),
            "multipassTimeNs": int(multipass_s * 1e9),
        }])
        return model

    def _model_class(self):
        raise NotImplementedError


class _VWModelBase(_VWParams, Model):
    weightsArrays = Param("weightsArrays", "weight + adaptive tables", None,
                          is_complex=True)

    def __init__(self, weights: Optional[np.ndarray] = None,
                 adaptive: Optional[np.ndarray] = None, **kwargs):
        super().__init__(**kwargs)
        if weights is not None:
            self.set("weightsArrays", {"weights": weights, "adaptive": adaptive})
        self._stats = None

    @property
    def weights(self) -> np.ndarray:
        return self.get("weightsArrays")["weights"]

    def getPerformanceStatistics(self) -> pd.DataFrame:
        """Per-rank perf stats DF (VowpalWabbitBase.scala:464-490)."""
        return self._stats if self._stats is not None else pd.DataFrame()

    def saveNativeModel(self, path: str):
        np.savez(path, **self.get("weightsArrays"))

    def getReadableModel(self) -> pd.DataFrame:
        w = self.weights
        nz = np.nonzero(w)[0]
        return pd.DataFrame({"index": nz, "weight": w[nz]})

    def _raw(self, df: pd.DataFrame) -> np.ndarray:
        device = default_device(self.get("device"))
        df, extra_cols = self._feature_frame(df)
        idx, val, off, _ = _extract_csr(df, self.get("featuresCol"),
                                        extra_cols, device)
        w = torch.from_numpy(self.weights).to(device)
        idx = idx & (w.numel() - 1)  # table mask, matching _fit
        return backend.vw_predict(idx, val, off, w).cpu().numpy()


@register
class VowpalWabbitRegressor(_VWBase):
    _default_loss = "squared"

    def _model_class(self):
        return VowpalWabbitRegressorModel


@register
class VowpalWabbitRegressorModel(_VWModelBase):
    def _transform(self, df: pd.DataFrame) -> pd.DataFrame:
        out = df.copy()
        out[self.get("predictionCol")] = self._raw(df).astype(np.float64)
        return out


@register
class VowpalWabbitClassifier(_VWBase):
    _default_loss = "logistic"
    _binary_labels = True
    rawPredictionCol = Param("rawPredictionCol", "margin column", "rawPrediction")
    probabilityCol = Param("probabilityCol", "probability column", "probability")
    labelConversion = Param("labelConversion", "convert {0,1} labels to "
                            "{-1,+1} before training "
                            "(VowpalWabbitClassifier.scala)", True, toBool)

    @property
    def _convert_labels(self):
        return self.get("labelConversion")

    def _fit(self, df):
        model = super()._fit(df)
        for p in ("rawPredictionCol", "probabilityCol", "labelConversion"):
            model.set(p, self.get(p))
        return model

    def _model_class(self):
        return VowpalWabbitClassificationModel


@register
class VowpalWabbitClassificationModel(_VWModelBase):
    rawPredictionCol = Param("rawPredictionCol", "margin column", "rawPrediction")
    probabilityCol = Param("probabilityCol", "probability column", "probability")
    labelConversion = Param("labelConversion", "labels were converted {0,1}→"
                            "{-1,+1} at fit (informational)", True, toBool)

    def _transform(self, df: pd.DataFrame) -> pd.DataFrame:
        raw = self._raw(df)
        p1 = 1.0 / (1.0 + np.exp(-raw))
        out = df.copy()
        out[self.get("rawPredictionCol")] = matrix_to_vector_column(
            np.stack([-raw, raw], axis=1))
        out[self.get("probabilityCol")] = matrix_to_vector_column(
            np.stack([1 - p1, p1], axis=1))
        out[self.get("predictionCol")] = (raw > 0).astype(np.float64)
        return out


# --------------------------------------------------------------- contextual bandit
class ContextualBanditMetrics:
    """IPS / SNIPS estimators (vw/.../VowpalWabbitContextualBandit.scala:53)."""

    def __init__(self):
        self.total = 0
        self.ips_num = 0.0
        self.snips_den = 0.0

    def add(self, prob_logged: float, cost: float, prob_pred_matches: float):
        self.total += 1
        w = prob_pred_matches / max(prob_logged, 1e-12)
        self.ips_num += w * cost
        self.snips_den += w

    @property
    def ips_estimate(self):
        return self.ips_num / max(self.total, 1)

    @property
    def snips_estimate(self):
        return self.ips_num / max(self.snips_den, 1e-12)


@register
class VowpalWabbitContextualBandit(_VWBase):
    """CB cost regression with IPS weighting: the chosen action's
    (shared ⊕ action ⊕ shared×action) features regress the observed cost with
    importance weight 1/p_logged (reference --cb_type ips semantics)."""
    _default_loss = "squared"
    sharedCol = Param("sharedCol", "shared-context sparse column", "shared")
    additionalSharedFeatures = Param(
        "additionalSharedFeatures", "extra shared-context columns, merged "
        "into the shared namespace with per-column murmur seeds "
        "(VowpalWabbitContextualBandit.scala:95)", None)
    featuresCol = Param("featuresCol", "per-action features (list of "
                        "SparseVector per row)", "features")
    chosenActionCol = Param("chosenActionCol", "1-based chosen action index",
                            "chosenAction")
    probabilityCol = Param("probabilityCol", "logged action probability",
                           "probability")
    labelCol = Param("labelCol", "observed cost", "cost")
    epsilon = Param("epsilon", "exploration for predicted policy", 0.05, toFloat)

    def _merged_shared(self, row) -> SparseVector:
        """sharedCol ⊕ additionalSharedFeatures, each extra column's indices
        offset by a murmur seed of its name (same namespace-seeding rule as
        _extract_csr)."""
        from .murmur import hash_string
        base = row[self.get("sharedCol")]
        extras = self.get("additionalSharedFeatures") or []
        if not extras:
            return base
        idx = [base.indices.astype(np.int64)]
        val = [base.values]
        for c in extras:
            v = row[c]
            seed = hash_string(c) & 0x3FFFFFFF
            idx.append((v.indices.astype(np.int64) + seed) & 0x3FFFFFFF)
            val.append(v.values)
        return SparseVector(1 << 30, np.concatenate(idx).astype(np.int32),
                            np.concatenate(val))

    def _combine(self, shared: SparseVector, action: SparseVector, mask: int):
        from .featurizer import FNV_PRIME
        si, sv = shared.indices.astype(np.int64), shared.values
        ai, av = action.indices.astype(np.int64), action.values
        cross_i = ((si[:, None] * FNV_PRIME) ^ ai[None, :]).reshape(-1)
        cross_v = (sv[:, None] * av[None, :]).reshape(-1)
        idx = np.concatenate([si & mask, ai & mask, cross_i & mask])
        val = np.concatenate([sv, av, cross_v])
        return idx.astype(np.int32), val.astype(np.float32)

    def _fit(self, df: pd.DataFrame):
        self._parse_args()
        device = default_device(self.get("device"))
        bits = self.get("numBits")
        tbl = 1 << bits
        mask = tbl - 1
        w = torch.zeros(tbl, dtype=torch.float32, device=device)
        g = torch.zeros(tbl, dtype=torch.float32, device=device)
        lr, l2, pt = self.get("learningRate"), self.get("l2"), self.get("powerT")

        idx_parts, val_parts, counts, labels, ws = [], [], [], [], []
        for _, row in df.iterrows():
            shared = self._merged_shared(row)
            actions = row[self.get("featuresCol")]
            chosen = int(row[self.get("chosenActionCol")]) - 1
            cost = float(row[self.get("labelCol")])
            prob = float(row[self.get("probabilityCol")])
            ia, va = self._combine(shared, actions[chosen], mask)
            idx_parts.append(ia)
            val_parts.append(va)
            counts.append(len(ia))
            labels.append(cost)
            ws.append(1.0 / max(prob, 1e-6))
        off = np.zeros(len(df) + 1, dtype=np.int64)
        np.cumsum(counts, out=off[1:])
        idx_t = torch.from_numpy(np.concatenate(idx_parts)).to(device)
        val_t = torch.from_numpy(np.concatenate(val_parts)).to(device)
        off_t = torch.from_numpy(off).to(device)
        y_t = torch.tensor(labels, dtype=torch.float32, device=device)
        w_t = torch.tensor(ws, dtype=torch.float32, device=device)

        bs = self.get("batchSize")
        n = len(df)
        for _ in range(self.get("numPasses")):
            for s in range(0, n, bs):
                e = min(s + bs, n)
                o = off_t[s:e + 1] - off_t[s]
                sl = slice(int(off_t[s]), int(off_t[e]))
                # IPS weights 1/prob can be huge → the invariant update is
                # what keeps a rare action's example from blowing up the step
                backend.vw_sgd_minibatch(idx_t[sl], val_t[sl], o, y_t[s:e],
                                         w, g, lr, l2, pt, "squared",
                                         w_t[s:e],
                                         invariant=self.get("invariant"))
        model = VowpalWabbitContextualBanditModel(weights=w.cpu().numpy(),
                                                  adaptive=g.cpu().numpy())
        for p in ("sharedCol", "additionalSharedFeatures", "featuresCol",
                  "predictionCol", "numBits", "epsilon"):
            model.set(p, self.get(p))
        return model


@register
class VowpalWabbitContextualBanditModel(_VWModelBase):
    sharedCol = Param("sharedCol", "shared-context sparse column", "shared")
    additionalSharedFeatures = Param(
        "additionalSharedFeatures", "extra shared-context columns", None)
    chosenActionCol = Param("chosenActionCol", "1-based chosen action index "
                            "column (for offline evaluation)", "chosenAction")
    probabilityCol = Param("probabilityCol", "logged action probability "
                           "column (for offline evaluation)", "probability")
    epsilon = Param("epsilon", "exploration rate", 0.05, toFloat)

    def _transform(self, df: pd.DataFrame) -> pd.DataFrame:
        est = VowpalWabbitContextualBandit()
        est.set("numBits", self.get("numBits"))
        est.set("sharedCol", self.get("sharedCol"))
        est.set("additionalSharedFeatures",
                self.get("additionalSharedFeatures"))
        mask = (1 << self.get("numBits")) - 1
        w = self.weights
        scores_col = []
        chosen_col = []
        for _, row in df.iterrows():
            shared = est._merged_shared(row)
            actions = row[self.get("featuresCol")]
            scores = []
            for a in actions:
                ia, va = est._combine(shared, a, mask)
                scores.append(float((w[ia] * va).sum()))
            scores_col.append(np.asarray(scores, dtype=np.float32))
            chosen_col.append(int(np.argmin(scores)) + 1)
        out = df.copy()
        out["predictedCosts"] = scores_col
        out[self.get("predictionCol")] = chosen_col
        # epsilon-greedy action distribution (VW --epsilon pmf: the greedy
        # action gets 1-eps+eps/K, every action eps/K)
        eps = float(self.get("epsilon"))
        probs_col = []
        for scores, chosen in zip(scores_col, chosen_col):
            K = max(len(scores), 1)
            p_ = np.full(K, eps / K, dtype=np.float32)
            p_[chosen - 1] += 1.0 - eps
            probs_col.append(p_)
        out["probabilities"] = probs_col
        return out
