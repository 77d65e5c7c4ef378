"""LightGBM-equivalent estimators: Classifier / Regressor / Ranker.

SparkML-shaped API parity with the reference learners
(lightgbm/src/main/scala/com/microsoft/ml/spark/lightgbm/LightGBMClassifier.scala:26,
LightGBMRegressor.scala, LightGBMRanker.scala:26) and the param surface of
params/LightGBMParams.scala — re-hosted on the MI355X trainer (trainer.py):
each data shard is a GPU rank, histogram sync is RCCL all_reduce over xGMI.
Scoring (transform) adds rawPrediction/probability/prediction columns plus
optional leafPrediction and featuresShap columns
(LightGBMClassifier.scala:111-160), batched through the HIP forest kernel —
never row-at-a-time JNI like the reference's UDF scoring path.
"""
from __future__ import annotations

from typing import Optional

import numpy as np
import pandas as pd
import torch

from ...core.param import Param, Params, toBool, toFloat, toInt, toString
from ...core.pipeline import Estimator, Model
from ...core.registry import register
from ...core.schema import (features_matrix, infer_feature_names,
                            matrix_to_vector_column)
from ...parallel.comm import get_comm
from ...utils.devices import default_device
from .booster import Booster
from .objectives import make_objective
from .trainer import TrainConfig, train_booster


class _GBDTParams(Params):
    """Shared LightGBM-style params (params/LightGBMParams.scala)."""

    labelCol = Param("labelCol", "label column", "label")
    featuresCol = Param("featuresCol", "features vector column", "features")
    featureCols = Param("featureCols", "list of numeric feature columns", None)
    weightCol = Param("weightCol", "row weight column", None)
    validationIndicatorCol = Param("validationIndicatorCol",
                                   "bool column marking validation rows", None)
    initScoreCol = Param("initScoreCol", "initial score column", None)
    predictionCol = Param("predictionCol", "prediction column", "prediction")

    numIterations = Param("numIterations", "number of boosting iterations", 100, toInt)
    learningRate = Param("learningRate", "shrinkage rate", 0.1, toFloat)
    numLeaves = Param("numLeaves", "max leaves per tree", 31, toInt)
    maxDepth = Param("maxDepth", "max tree depth (-1 = unlimited)", -1, toInt)
    maxBin = Param("maxBin", "max number of feature bins", 255, toInt)
    lambdaL1 = Param("lambdaL1", "L1 regularization", 0.0, toFloat)
    lambdaL2 = Param("lambdaL2", "L2 regularization", 0.0, toFloat)
    minDataInLeaf = Param("minDataInLeaf", "min rows per leaf", 20, toInt)
    minSumHessianInLeaf = Param("minSumHessianInLeaf", "min hessian per leaf",
                                1e-3, toFloat)
    minGainToSplit = Param("minGainToSplit", "min split gain", 0.0, toFloat)
    featureFraction = Param("featureFraction", "feature subsample per tree", 1.0, toFloat)
    baggingFraction = Param("baggingFraction", "row subsample fraction", 1.0, toFloat)
    baggingFreq = Param("baggingFreq", "bagging frequency (0=off)", 0, toInt)
    baggingSeed = Param("baggingSeed", "bagging seed", 3, toInt)
    boostingType = Param("boostingType", "gbdt|rf|dart|goss", "gbdt", toString)
    topRate = Param("topRate", "GOSS large-gradient keep rate", 0.2, toFloat)
    otherRate = Param("otherRate", "GOSS small-gradient sample rate", 0.1, toFloat)
    dropRate = Param("dropRate", "DART drop rate", 0.1, toFloat)
    skipDrop = Param("skipDrop", "DART skip probability", 0.5, toFloat)
    maxDrop = Param("maxDrop", "DART max dropped trees", 50, toInt)
    maxDeltaStep = Param("maxDeltaStep", "max leaf output", 0.0, toFloat)
    earlyStoppingRound = Param("earlyStoppingRound", "early stop patience (0=off)",
                               0, toInt)
    objective = Param("objective", "objective name", None)
    metric = Param("metric", "evaluation metric", "", toString)
    seed = Param("seed", "random seed", 0, toInt)
    numBatches = Param("numBatches", "split data into n sequential training batches "
                       "(LightGBMBase.scala:46-61)", 0, toInt)
    verbosity = Param("verbosity", "log verbosity", -1, toInt)
    isProvideTrainingMetric = Param("isProvideTrainingMetric",
                                    "log train metrics per iteration", False, toBool)
    useBarrierExecutionMode = Param("useBarrierExecutionMode",
                                    "gang-schedule ranks (no-op: ranks are always "
                                    "gang-launched here)", False, toBool)
    parallelism = Param("parallelism", "data_parallel|voting_parallel", "data_parallel")
    topK = Param("topK", "voting-parallel top-K", 20, toInt)
    categoricalSlotIndexes = Param("categoricalSlotIndexes",
                                   "indexes of categorical features", None)
    categoricalSlotNames = Param("categoricalSlotNames",
                                 "names of categorical features (resolved "
                                 "against feature names)", None)
    slotNames = Param("slotNames", "override feature slot names", None)
    modelString = Param("modelString", "warm-start model text", "", toString)
    lightGBMBooster = Param("lightGBMBooster", "warm-start Booster object "
                            "(LightGBMParams.scala lightGBMBooster)", None,
                            is_complex=True)
    fobj = Param("fobj", "custom objective callable (preds, label, weight) "
                 "-> (grad, hess) — FObjTrait analog", None, is_complex=True)
    boostFromAverage = Param("boostFromAverage", "start boosting from the "
                             "global label mean", True, toBool)
    improvementTolerance = Param("improvementTolerance",
                                 "min metric delta that counts as an early-"
                                 "stopping improvement", 0.0, toFloat)
    posBaggingFraction = Param("posBaggingFraction",
                               "positive-row bagging fraction", 1.0, toFloat)
    negBaggingFraction = Param("negBaggingFraction",
                               "negative-row bagging fraction", 1.0, toFloat)
    binSampleCount = Param("binSampleCount", "rows sampled for quantile bin "
                           "boundaries", 200000, toInt)
    maxBinByFeature = Param("maxBinByFeature", "per-feature max bin caps", None)
    uniformDrop = Param("uniformDrop", "DART: uniform drop selection", True,
                        toBool)
    xgboostDartMode = Param("xgboostDartMode", "DART: xgboost normalization "
                            "mode", False, toBool)
    startIteration = Param("startIteration", "first iteration used at "
                           "predict time", 0, toInt)
    leafPredictionCol = Param("leafPredictionCol",
                              "output column for per-tree leaf indices "
                              "(propagated to the model)", None)
    featuresShapCol = Param("featuresShapCol",
                            "output column for SHAP contributions "
                            "(propagated to the model)", None)
    # Spark/JVM topology params accepted for API compatibility; topology is
    # one-process-per-GPU with RCCL here, so they change nothing (documented
    # in PARITY.md — LightGBMParams.scala:54-100)
    chunkSize = Param("chunkSize", "ingestion chunk size (obsolete: data is "
                      "device-resident)", 10000, toInt)
    defaultListenPort = Param("defaultListenPort", "obsolete (RCCL "
                              "rendezvous via MASTER_ADDR)", 12400, toInt)
    driverListenPort = Param("driverListenPort", "obsolete", 0, toInt)
    timeout = Param("timeout", "training timeout seconds (driver-level here)",
                    1200.0, toFloat)
    numTasks = Param("numTasks", "obsolete: world size = launched ranks", 0,
                     toInt)
    numThreads = Param("numThreads", "obsolete: GPU kernels replace the "
                       "OpenMP pool", 0, toInt)
    useSingleDatasetMode = Param("useSingleDatasetMode", "obsolete: one "
                                 "process per GPU owns its shard", False,
                                 toBool)
    matrixType = Param("matrixType", "auto|dense|sparse — sparse trains from "
                       "binned CSR without densifying (auto samples the "
                       "column like DatasetUtils.scala:49)", "auto", toString)
    checkpointDir = Param("checkpointDir", "directory for iteration-level "
                          "checkpoints (elastic restart resumes from it)",
                          None)
    checkpointInterval = Param("checkpointInterval", "write a checkpoint "
                               "every k iterations (0 = off)", 0, toInt)
    device = Param("device", "cpu|cuda|auto", "auto", toString)

    def _train_config(self) -> TrainConfig:
        return TrainConfig(
            num_iterations=self.get("numIterations"),
            learning_rate=self.get("learningRate"),
            num_leaves=self.get("numLeaves"),
            max_depth=self.get("maxDepth"),
            max_bin=min(self.get("maxBin"), 255),
            lambda_l1=self.get("lambdaL1"),
            lambda_l2=self.get("lambdaL2"),
            min_data_in_leaf=self.get("minDataInLeaf"),
            min_sum_hessian_in_leaf=self.get("minSumHessianInLeaf"),
            min_gain_to_split=self.get("minGainToSplit"),
            feature_fraction=self.get("featureFraction"),
            bagging_fraction=self.get("baggingFraction"),
            bagging_freq=self.get("baggingFreq"),
            bagging_seed=self.get("baggingSeed"),
            boosting=self.get("boostingType"),
            top_rate=self.get("topRate"),
            other_rate=self.get("otherRate"),
            drop_rate=self.get("dropRate"),
            skip_drop=self.get("skipDrop"),
            max_drop=self.get("maxDrop"),
            max_delta_step=self.get("maxDeltaStep"),
            seed=self.get("seed"),
            early_stopping_round=self.get("earlyStoppingRound"),
            is_provide_training_metric=self.get("isProvideTrainingMetric"),
            metric=self.get("metric"),
            verbosity=self.get("verbosity"),
            categorical_features=self.get("categoricalSlotIndexes"),
            parallelism=self.get("parallelism"),
            top_k=self.get("topK"),
            boost_from_average=self.get("boostFromAverage"),
            improvement_tolerance=self.get("improvementTolerance"),
            pos_bagging_fraction=self.get("posBaggingFraction"),
            neg_bagging_fraction=self.get("negBaggingFraction"),
            bin_sample_count=self.get("binSampleCount"),
            max_bin_by_feature=self.get("maxBinByFeature"),
            uniform_drop=self.get("uniformDrop"),
            xgboost_dart_mode=self.get("xgboostDartMode"),
        )

    def _device(self):
        return default_device(self.get("device"))


class _GBDTEstimatorBase(_GBDTParams, Estimator):
    _default_objective = "regression"

    def _make_objective(self, y: torch.Tensor):
        name = self.get("objective") or self._default_objective
        num_class = int(y.max().item()) + 1 if name in ("multiclass", "softmax") else 2
        return make_objective(name, num_class=num_class,
                              fobj=self.get("fobj"))

    def _use_sparse(self, df: pd.DataFrame) -> bool:
        """matrixType='sparse' forces CSR; 'auto' samples the column like the
        reference (DatasetUtils.scala:49 sampleRowsForArrayType)."""
        from .sparse import looks_sparse
        if self.get("featureCols"):
            return False
        fc = self.get("featuresCol")
        mt = (self.get("matrixType") or "auto").lower()
        if mt == "dense" or fc not in df.columns or not len(df):
            return False
        return mt == "sparse" or looks_sparse(df[fc])

    def _extract(self, df: pd.DataFrame, device):
        y = df[self.get("labelCol")].to_numpy(dtype=np.float32)
        yt = torch.from_numpy(y).to(device)
        w = None
        if self.get("weightCol"):
            w = torch.from_numpy(
                df[self.get("weightCol")].to_numpy(dtype=np.float32)).to(device)
        if self._use_sparse(df):
            from .sparse import CsrMatrix
            Xs = CsrMatrix.from_sparse_vectors(
                df[self.get("featuresCol")]).to(device)
            return Xs, yt, w
        X = features_matrix(df, self.get("featuresCol"), self.get("featureCols"))
        Xt = torch.from_numpy(np.ascontiguousarray(X)).to(device)
        return Xt, yt, w

    def _fit(self, df: pd.DataFrame):
        device = self._device()
        comm = get_comm()
        names = infer_feature_names(df, self.get("featuresCol"),
                                    self.get("featureCols"))
        if self.get("slotNames"):  # slotNames overrides inferred names
            sn = list(self.get("slotNames"))
            names = sn + names[len(sn):]
        csn = self.get("categoricalSlotNames")
        if csn:  # resolve categorical names -> slot indexes
            cat = list(self.get("categoricalSlotIndexes") or [])
            cat += [names.index(nm) for nm in csn if nm in names]
            self.set("categoricalSlotIndexes", sorted(set(cat)))
        valid_df = None
        vic = self.get("validationIndicatorCol")
        if vic and vic in df.columns:
            valid_df = df[df[vic].astype(bool)]
            df = df[~df[vic].astype(bool)]

        n_batches = self.get("numBatches") or 0
        batches = ([df] if n_batches <= 1 else
                   np.array_split(np.arange(len(df)), n_batches))

        init = None
        ms = self.get("modelString")
        if ms:
            init = Booster.load_from_string(ms)
        elif self.get("lightGBMBooster") is not None:
            init = self.get("lightGBMBooster")  # warm start from object

        valid_sets = None
        if valid_df is not None and len(valid_df):
            Xv, yv, wv = self._extract(valid_df, device)
            valid_sets = [(Xv, yv, wv)]
            valid_groups = self._group_sizes(valid_df, device)

        from .metrics import default_metrics_fn
        booster = init
        stats = None
        for bi, b in enumerate(batches):
            part = df if n_batches <= 1 else df.iloc[b]
            # each sequential batch gets its own checkpoint namespace —
            # otherwise batch 2+ would "resume" from batch 1's completed
            # checkpoint and train nothing
            ckdir = self.get("checkpointDir")
            if ckdir and n_batches > 1:
                import os as _os
                ckdir = _os.path.join(ckdir, f"batch{bi}")
            Xt, yt, w = self._extract(part, device)
            w = self._adjust_weights(yt, w)
            group = self._group_sizes(part, device)
            init_score = None
            isc = self.get("initScoreCol")
            if isc and isc in part.columns:
                init_score = torch.from_numpy(
                    part[isc].to_numpy(dtype=np.float32)).to(device)
            booster, stats = train_booster(
                Xt, yt, self._train_config(), self._make_objective(yt), comm,
                weight=w, group_sizes=group, feature_names=names,
                valid_sets=valid_sets, init_booster=booster,
                metrics_fn=self._metrics_fn(
                    valid_groups if valid_sets else None),
                init_score=init_score,
                checkpoint_dir=ckdir,
                checkpoint_every=self.get("checkpointInterval"))
        model = self._model_class()(booster=booster)
        for p in ("labelCol", "featuresCol", "featureCols", "predictionCol",
                  "leafPredictionCol", "featuresShapCol"):
            model.set(p, self.get(p))
        model._training_stats = stats
        return model

    def _group_sizes(self, df, device):
        return None

    def _metrics_fn(self, valid_groups=None):
        from .metrics import default_metrics_fn
        return default_metrics_fn(self.get("metric") or None)

    def _adjust_weights(self, yt, w):
        return w

    def _model_class(self):
        raise NotImplementedError


class _GBDTModelBase(Model):
    labelCol = Param("labelCol", "label column", "label")
    featuresCol = Param("featuresCol", "features vector column", "features")
    featureCols = Param("featureCols", "list of numeric feature columns", None)
    predictionCol = Param("predictionCol", "prediction column", "prediction")
    leafPredictionCol = Param("leafPredictionCol",
                              "output column for per-tree leaf indices", None)
    featuresShapCol = Param("featuresShapCol",
                            "output column for SHAP contributions", None)
    boosterModelStr = Param("boosterModelStr", "serialized booster", None,
                            is_complex=True)
    startIteration = Param("startIteration", "first iteration used at "
                           "predict time", 0, toInt)
    numIterations = Param("numIterations", "iterations used at predict time "
                          "(-1 = best/all)", -1, toInt)
    device = Param("device", "cpu|cuda|auto", "auto", toString)

    def __init__(self, booster: Optional[Booster] = None, **kwargs):
        super().__init__(**kwargs)
        self._booster = booster
        if booster is not None:
            self.set("boosterModelStr", booster.save_to_string())

    def _post_deserialize_init(self):
        s = self.get("boosterModelStr")
        self._booster = Booster.load_from_string(s) if s else None

    @property
    def booster(self) -> Booster:
        if getattr(self, "_booster", None) is None:
            self._post_deserialize_init()
        return self._booster

    def getNativeModel(self) -> str:
        return self.booster.save_to_string()

    @classmethod
    def loadNativeModelFromString(cls, s: str, **kwargs):
        """Build a scoring model from a model string — ours (JSON) or stock
        LightGBM native text (LightGBMClassificationModel.
        loadNativeModelFromString, LightGBMClassifier.scala:247)."""
        return cls(booster=Booster.load_from_string(s), **kwargs)

    @classmethod
    def loadNativeModelFromFile(cls, path: str, **kwargs):
        """Analog of loadNativeModelFromFile (LightGBMClassifier.scala:240)."""
        with open(path) as f:
            return cls(booster=Booster.load_from_string(f.read()), **kwargs)

    def saveNativeModel(self, path: str):
        """Analog of saveNativeModel (LightGBMClassifier.scala:185-205)."""
        with open(path, "w") as f:
            f.write(self.booster.save_to_string())

    def getFeatureImportances(self, importance_type: str = "split"):
        return self.booster.feature_importances(importance_type).tolist()

    def score_matrix(self, X: np.ndarray, target_col: str, classes):
        """Explainer fast path (LocalExplainer._score_matrix): raw feature
        matrix → target-column values with ONE forest pass and no DataFrame
        construction — the perturbation fan-out scores millions of rows."""
        device = default_device(self.get("device"))
        Xt = torch.from_numpy(
            np.ascontiguousarray(X, dtype=np.float32)).to(device)
        b = self.booster
        raw = b.predict_raw(Xt, self.get("startIteration"),
                            self.get("numIterations"))
        params = self.params()
        prob_col = self.get("probabilityCol") if "probabilityCol" in params \
            else None
        raw_col = self.get("rawPredictionCol") if "rawPredictionCol" in params \
            else None
        if target_col == prob_col:
            if b.objective == "binary":
                p1 = torch.sigmoid(b.sigmoid * raw)
                vals = torch.cat([1 - p1, p1], dim=-1)
            elif b.objective in ("multiclass", "softmax"):
                vals = torch.softmax(raw, dim=-1)
            else:
                vals = raw
        elif target_col == raw_col:
            vals = torch.cat([-raw, raw], dim=-1) if b.objective == "binary" \
                else raw
        elif target_col == self.get("predictionCol"):
            if b.objective == "binary":
                vals = (torch.sigmoid(b.sigmoid * raw) > 0.5).float()
            elif b.objective in ("multiclass", "softmax"):
                vals = raw.argmax(dim=-1, keepdim=True).float()
            else:
                vals = raw
        else:  # unknown target column: the caller falls back to transform()
            return None
        v = vals.cpu().numpy().astype(np.float64)
        if v.shape[1] == 1:  # scalar target (regression/prediction)
            return v
        return v[:, list(classes)]

    def _X(self, df: pd.DataFrame, device):
        from .sparse import CsrMatrix, looks_sparse
        fc = self.get("featuresCol")
        if (not self.get("featureCols") and fc in df.columns and len(df)
                and looks_sparse(df[fc])):
            return CsrMatrix.from_sparse_vectors(df[fc]).to(device)
        X = features_matrix(df, self.get("featuresCol"), self.get("featureCols"))
        return torch.from_numpy(np.ascontiguousarray(X)).to(device)

    def _maybe_extra_cols(self, df, out, X):
        if self.get("leafPredictionCol"):
            leaves = self.booster.predict_leaf(X).cpu().numpy().astype(np.float64)
            out[self.get("leafPredictionCol")] = matrix_to_vector_column(leaves)
        if self.get("featuresShapCol"):
            contrib = self.booster.predict_contrib(X)
            out[self.get("featuresShapCol")] = matrix_to_vector_column(contrib)
        return out


@register
class LightGBMClassifier(_GBDTEstimatorBase):
    """Binary/multiclass GBDT classifier (LightGBMClassifier.scala:26)."""
    _default_objective = "binary"
    rawPredictionCol = Param("rawPredictionCol", "raw margin column", "rawPrediction")
    probabilityCol = Param("probabilityCol", "probability column", "probability")
    isUnbalance = Param("isUnbalance", "re-weight positives by n_neg/n_pos "
                        "(binary only)", False, toBool)

    def getActualNumClasses(self) -> int:
        """Number of classes inferred at fit time (LightGBMClassifier
        actualNumClasses); 2 until fit."""
        return getattr(self, "_actual_num_classes", 2)

    def _adjust_weights(self, yt, w):
        if not self.get("isUnbalance"):
            return w
        pos = yt > 0
        n_pos = float(pos.sum())
        n_neg = float(yt.numel() - n_pos)
        if n_pos == 0 or n_neg == 0:
            return w
        scale = torch.where(pos, n_neg / n_pos, 1.0)
        return scale if w is None else w * scale

    def _model_class(self):
        return LightGBMClassificationModel

    def _fit(self, df):
        model = super()._fit(df)
        for p in ("rawPredictionCol", "probabilityCol", "isUnbalance"):
            model.set(p, self.get(p))
        return model


@register
class LightGBMClassificationModel(_GBDTModelBase):
    rawPredictionCol = Param("rawPredictionCol", "raw margin column", "rawPrediction")
    probabilityCol = Param("probabilityCol", "probability column", "probability")
    thresholds = Param("thresholds", "per-class prediction thresholds", None)
    isUnbalance = Param("isUnbalance", "was the estimator fit with unbalance "
                        "re-weighting (informational)", False, toBool)

    def getActualNumClasses(self) -> int:
        """LightGBMClassificationModel.getActualNumClasses: classes in the
        trained booster (1 output plane = binary)."""
        n = self.booster.num_planes if hasattr(self.booster, "num_planes")             else 1
        return 2 if n <= 1 else n

    def _transform(self, df: pd.DataFrame) -> pd.DataFrame:
        device = default_device(self.get("device"))
        X = self._X(df, device)
        b = self.booster
        si, ni = self.get("startIteration"), self.get("numIterations")
        raw = b.predict_raw(X, si, ni)
        # probability from the raw margins already computed — traversing
        # the forest once, not twice (this path is the explainer fan-out's
        # inner loop)
        if b.objective == "binary":
            p1 = torch.sigmoid(b.sigmoid * raw)
            prob = torch.cat([1 - p1, p1], dim=-1)
        elif b.objective in ("multiclass", "softmax"):
            prob = torch.softmax(raw, dim=-1)
        else:
            prob = raw
        out = df.copy()
        if b.objective == "binary":
            raw2 = torch.cat([-raw, raw], dim=-1)
        else:
            raw2 = raw
        out[self.get("rawPredictionCol")] = matrix_to_vector_column(
            raw2.cpu().numpy())
        out[self.get("probabilityCol")] = matrix_to_vector_column(
            prob.cpu().numpy())
        th = self.get("thresholds")
        if th:
            scaled = prob.cpu().numpy() / np.asarray(th, dtype=np.float32)
            pred = scaled.argmax(axis=1).astype(np.float64)
        else:
            pred = prob.argmax(dim=-1).cpu().numpy().astype(np.float64)
        out[self.get("predictionCol")] = pred
        return self._maybe_extra_cols(df, out, X)

    @property
    def numClasses(self):
        return 2 if self.booster.objective == "binary" else self.booster.n_outputs


@register
class LightGBMRegressor(_GBDTEstimatorBase):
    """GBDT regressor (LightGBMRegressor.scala)."""
    _default_objective = "regression"
    alpha = Param("alpha", "huber/quantile alpha", 0.9, toFloat)
    tweedieVariancePower = Param("tweedieVariancePower", "tweedie rho", 1.5, toFloat)

    def _fit(self, df):
        model = super()._fit(df)
        for p in ("alpha", "tweedieVariancePower"):
            model.set(p, self.get(p))
        return model

    def _make_objective(self, y):
        name = self.get("objective") or "regression"
        return make_objective(name, alpha=self.get("alpha"),
                              tweedie_variance_power=self.get("tweedieVariancePower"))

    def _model_class(self):
        return LightGBMRegressionModel


@register
class LightGBMRegressionModel(_GBDTModelBase):
    alpha = Param("alpha", "huber/quantile alpha used at fit", 0.9, toFloat)
    tweedieVariancePower = Param("tweedieVariancePower",
                                 "tweedie rho used at fit", 1.5, toFloat)

    def _transform(self, df: pd.DataFrame) -> pd.DataFrame:
        device = default_device(self.get("device"))
        X = self._X(df, device)
        raw = self.booster.predict_raw(X, self.get("startIteration"),
                                       self.get("numIterations"))
        if self.booster.objective in ("poisson", "tweedie"):
            raw = torch.exp(raw)
        out = df.copy()
        out[self.get("predictionCol")] = raw.squeeze(-1).cpu().numpy().astype(np.float64)
        return self._maybe_extra_cols(df, out, X)


@register
class LightGBMRanker(_GBDTEstimatorBase):
    """LambdaRank ranker (LightGBMRanker.scala:26; groupCol/labelGain/evalAt
    params :36-52)."""
    _default_objective = "lambdarank"
    groupCol = Param("groupCol", "query group column", "group")
    labelGain = Param("labelGain", "per-label gain table", None)
    maxPosition = Param("maxPosition", "NDCG truncation", 10, toInt)
    evalAt = Param("evalAt", "NDCG eval positions", None)
    repartitionByGroupingColumn = Param(
        "repartitionByGroupingColumn", "make each query group contiguous "
        "before training (LightGBMParams.scala:76); False trusts the input "
        "ordering", True, toBool)

    def _make_objective(self, y):
        return make_objective("lambdarank", label_gain=self.get("labelGain"))

    def _group_sizes(self, df, device):
        gc = self.get("groupCol")
        sizes = df.groupby(gc, sort=False).size().to_numpy()
        return torch.from_numpy(sizes.astype(np.int64))

    def _fit(self, df):
        # rows of one query group must be contiguous (reference repartitions by
        # grouping column, LightGBMParams.scala:76) — sort locally by group
        if self.get("repartitionByGroupingColumn"):
            df = df.sort_values(self.get("groupCol"),
                                kind="stable").reset_index(drop=True)
        model = super()._fit(df)
        for p in ("labelGain", "maxPosition", "evalAt"):
            model.set(p, self.get(p))
        return model

    def _metrics_fn(self, valid_groups=None):
        """Group-aware NDCG@k on the validation split (evalAt /
        maxPosition, labelGain semantics — LightGBMRanker.scala:36-52);
        higher-is-better early stopping keys off the 'ndcg@k' name."""
        eval_at = list(self.get("evalAt") or [self.get("maxPosition")])
        gains_tbl = self.get("labelGain")
        sizes = (valid_groups.tolist() if valid_groups is not None else None)

        def fn(booster, Xv, yv, wv, comm):
            raw = booster.predict_raw(Xv).squeeze(-1)
            scores = raw.cpu().numpy()
            y = yv.cpu().numpy()
            group_sizes = sizes if sizes else [len(y)]
            out = {}
            for k in eval_at:
                vals = []
                start = 0
                for sz in group_sizes:
                    sl = slice(start, start + int(sz))
                    start += int(sz)
                    ys, ss = y[sl], scores[sl]
                    if len(ys) == 0:
                        continue
                    order = np.argsort(-ss, kind="stable")
                    g = (np.asarray(gains_tbl)[ys.astype(int)]
                         if gains_tbl is not None else 2.0 ** ys - 1.0)
                    disc = 1.0 / np.log2(np.arange(2, len(ys) + 2))
                    kk = min(int(k), len(ys))
                    dcg = float((g[order][:kk] * disc[:kk]).sum())
                    idcg = float((np.sort(g)[::-1][:kk] * disc[:kk]).sum())
                    vals.append(dcg / idcg if idcg > 0 else 0.0)
                m = float(np.mean(vals)) if vals else 0.0
                t = torch.tensor([m, 1.0], device=Xv.device)
                comm.all_reduce(t)
                out[f"ndcg@{int(k)}"] = float(t[0] / t[1])
            return out

        return fn

    def _model_class(self):
        return LightGBMRankerModel


@register
class LightGBMRankerModel(_GBDTModelBase):
    labelGain = Param("labelGain", "per-label gain table used at fit", None)
    maxPosition = Param("maxPosition", "NDCG truncation used at fit", 10,
                        toInt)
    evalAt = Param("evalAt", "NDCG eval positions used at fit", None)

    def _transform(self, df: pd.DataFrame) -> pd.DataFrame:
        device = default_device(self.get("device"))
        X = self._X(df, device)
        raw = self.booster.predict_raw(X, self.get("startIteration"),
                                       self.get("numIterations"))
        out = df.copy()
        out[self.get("predictionCol")] = raw.squeeze(-1).cpu().numpy().astype(np.float64)
        return self._maybe_extra_cols(df, out, X)
