"""SAR (Smart Adaptive Recommendations) + indexer + ranking evaluator.

Parity: core/.../recommendation/SAR.scala:36-209 (item-item co-occurrence
similarity with jaccard/lift, time-decayed user-item affinity,
score = affinity × similarity), SARModel:22 (recommendForAllUsers),
RecommendationIndexer, RankingEvaluator (NDCG/MAP/precision@k).
Dense torch matmuls run on the matrix cores via rocBLAS."""
from __future__ import annotations

import math

import numpy as np
import pandas as pd
import torch

from ..core.param import Param, toFloat, toInt, toString
from ..core.pipeline import Estimator, Model, Transformer
from ..core.registry import register
from ..utils.devices import default_device


@register
class RecommendationIndexer(Estimator):
    userInputCol = Param("userInputCol", "raw user column", "user")
    itemInputCol = Param("itemInputCol", "raw item column", "item")
    userOutputCol = Param("userOutputCol", "indexed user column", "userIdx")
    itemOutputCol = Param("itemOutputCol", "indexed item column", "itemIdx")

    def _fit(self, df: pd.DataFrame):
        users = {u: i for i, u in enumerate(pd.unique(df[self.get("userInputCol")]))}
        items = {v: i for i, v in enumerate(pd.unique(df[self.get("itemInputCol")]))}
        m = RecommendationIndexerModel()
        m.set("userMap", {str(k): v for k, v in users.items()})
        m.set("itemMap", {str(k): v for k, v in items.items()})
        for p in ("userInputCol", "itemInputCol", "userOutputCol",
                  "itemOutputCol"):
            m.set(p, self.get(p))
        return m


@register
class RecommendationIndexerModel(Model):
    userInputCol = Param("userInputCol", "raw user column", "user")
    itemInputCol = Param("itemInputCol", "raw item column", "item")
    userOutputCol = Param("userOutputCol", "indexed user column", "userIdx")
    itemOutputCol = Param("itemOutputCol", "indexed item column", "itemIdx")
    userMap = Param("userMap", "user → index", None, is_complex=True)
    itemMap = Param("itemMap", "item → index", None, is_complex=True)

    def _transform(self, df: pd.DataFrame) -> pd.DataFrame:
        out = df.copy()
        um, im = self.get("userMap"), self.get("itemMap")
        out[self.get("userOutputCol")] = [
            um.get(str(u), -1) for u in df[self.get("userInputCol")]]
        out[self.get("itemOutputCol")] = [
            im.get(str(v), -1) for v in df[self.get("itemInputCol")]]
        return out

    def recoverUser(self):
        return {v: k for k, v in self.get("userMap").items()}

    def recoverItem(self):
        return {v: k for k, v in self.get("itemMap").items()}


@register
class SAR(Estimator):
    userCol = Param("userCol", "user index column", "userIdx")
    itemCol = Param("itemCol", "item index column", "itemIdx")
    ratingCol = Param("ratingCol", "rating column", "rating")
    timeCol = Param("timeCol", "timestamp column (seconds)", None)
    supportThreshold = Param("supportThreshold", "min co-occurrence", 4, toInt)
    similarityFunction = Param("similarityFunction",
                               "jaccard|lift|cooccurrence", "jaccard", toString)
    timeDecayCoeff = Param("timeDecayCoeff", "affinity half-life (days)", 30,
                           toInt)
    startTime = Param("startTime", "custom 'now' reference time for the "
                      "decay when scoring historical data (SAR.scala:242)",
                      None)
    startTimeFormat = Param("startTimeFormat", "strftime format of startTime "
                            "(SAR.scala:50)", "EEE MMM dd HH:mm:ss Z yyyy")
    activityTimeFormat = Param("activityTimeFormat", "strftime format of a "
                               "string timeCol (SAR.scala:53)",
                               "yyyy/MM/dd'T'h:mm:ss")

    def _fit(self, df: pd.DataFrame):
        device = default_device("auto")
        u = torch.from_numpy(df[self.get("userCol")].to_numpy(np.int64))
        i = torch.from_numpy(df[self.get("itemCol")].to_numpy(np.int64))
        r = torch.from_numpy(df[self.get("ratingCol")].to_numpy(np.float32)) \
            if self.get("ratingCol") in df.columns else torch.ones(len(df))
        n_users = int(u.max()) + 1 if len(u) else 0
        n_items = int(i.max()) + 1 if len(i) else 0

        # time-decayed affinity (SAR.scala:80-130): rating * 2^(-(t_ref - t)/T)
        if self.get("timeCol") and self.get("timeCol") in df.columns:
            tv = df[self.get("timeCol")]
            if tv.dtype == object or str(tv.dtype).startswith("datetime"):
                # string/datetime timestamps (activityTimeFormat semantics —
                # pandas infers the format; epoch seconds)
                tv = pd.to_datetime(tv).astype("int64") / 1e9
                t = torch.from_numpy(tv.to_numpy(np.float64))
            else:
                t = torch.from_numpy(tv.to_numpy(np.float64))
            st = self.get("startTime")
            t_ref = (float(pd.Timestamp(st).timestamp()) if st
                     else float(t.max()))
            half_life = self.get("timeDecayCoeff") * 86400.0
            decay = torch.pow(2.0, -(t_ref - t) / half_life).float()
            r = r * decay
        A = torch.zeros(n_users, n_items, device=device)
        A[u.to(device), i.to(device)] += r.to(device)

        # item-item co-occurrence on binarized interactions
        B = (A > 0).float()
        C = B.t() @ B  # co-occurrence counts
        thr = float(self.get("supportThreshold"))
        C = torch.where(C >= thr, C, torch.zeros_like(C))
        diag = C.diagonal().clamp_min(1e-12)
        sim_fn = self.get("similarityFunction")
        if sim_fn == "jaccard":
            S = C / (diag.unsqueeze(0) + diag.unsqueeze(1) - C).clamp_min(1e-12)
        elif sim_fn == "lift":
            S = C / (diag.unsqueeze(0) * diag.unsqueeze(1))
        else:
            S = C
        S = torch.where(C > 0, S, torch.zeros_like(S))

        model = SARModel()
        model.set("sarArrays", {"affinity": A.cpu().numpy(),
                                "similarity": S.cpu().numpy()})
        for p in ("userCol", "itemCol", "ratingCol"):
            model.set(p, self.get(p))
        return model


@register
class SARModel(Model):
    userCol = Param("userCol", "user index column", "userIdx")
    itemCol = Param("itemCol", "item index column", "itemIdx")
    ratingCol = Param("ratingCol", "rating column", "rating")
    predictionCol = Param("predictionCol", "score column", "prediction")
    sarArrays = Param("sarArrays", "affinity + similarity matrices", None,
                      is_complex=True)

    def getItemDataFrame(self) -> pd.DataFrame:
        """Item-item similarity as a DataFrame (SARModel itemDataFrame):
        one row per item, `similarity` = that item's row vector."""
        S = self.get("sarArrays")["similarity"]
        return pd.DataFrame({"itemID": np.arange(len(S)),
                             "similarity": list(S.astype(np.float64))})

    def getUserDataFrame(self) -> pd.DataFrame:
        """User-affinity as a DataFrame (SARModel userDataFrame)."""
        A = self.get("sarArrays")["affinity"]
        return pd.DataFrame({"userID": np.arange(len(A)),
                             "affinity": list(A.astype(np.float64))})

    def _mats(self):
        d = self.get("sarArrays")
        device = default_device("auto")
        return (torch.from_numpy(d["affinity"]).to(device),
                torch.from_numpy(d["similarity"]).to(device))

    def _transform(self, df: pd.DataFrame) -> pd.DataFrame:
        """Score each (user, item) row: affinity[u] · similarity[:, i]."""
        A, S = self._mats()
        scores = A @ S  # (users, items)
        u = df[self.get("userCol")].to_numpy(np.int64)
        i = df[self.get("itemCol")].to_numpy(np.int64)
        out = df.copy()
        sc = scores.cpu().numpy()
        out[self.get("predictionCol")] = [
            float(sc[a, b]) if 0 <= a < sc.shape[0] and 0 <= b < sc.shape[1]
            else 0.0 for a, b in zip(u, i)]
        return out

    def recommendForAllUsers(self, k: int, remove_seen: bool = True
                             ) -> pd.DataFrame:
        A, S = self._mats()
        scores = A @ S
        if remove_seen:
            scores = torch.where(A > 0, torch.full_like(scores, -1e30), scores)
        vals, idx = torch.topk(scores, min(k, scores.shape[1]), dim=1)
        # ONE device→host transfer (a per-user .cpu() was a sync per row)
        vals_np = vals.cpu().numpy()
        idx_np = idx.cpu().numpy()
        ucol, icol = self.get("userCol"), self.get("itemCol")
        rows = []
        for uidx in range(vals_np.shape[0]):
            rows.append({
                ucol: uidx,
                "recommendations": [
                    {icol: int(j), "rating": float(v)}
                    for j, v in zip(idx_np[uidx], vals_np[uidx])
                    if v > -1e29]})
        return pd.DataFrame(rows)


@register
class RankingEvaluator(Transformer):
    """NDCG@k / MAP@k / precision@k / recall@k over (predictions, labels)
    list columns (core/.../recommendation/RankingEvaluator parity)."""
    k = Param("k", "cutoff", 10, toInt)
    metricName = Param("metricName", "ndcgAt|map|precisionAtk|recallAtK",
                       "ndcgAt", toString)
    predictionCol = Param("predictionCol", "ranked item list column",
                          "prediction")
    labelCol = Param("labelCol", "relevant item list column", "label")

    def evaluate(self, df: pd.DataFrame) -> float:
        k = self.get("k")
        metric = self.get("metricName")
        vals = []
        for _, row in df.iterrows():
            pred = list(row[self.get("predictionCol")])[:k]
            truth = set(row[self.get("labelCol")])
            if not truth:
                continue
            hits = [1.0 if p in truth else 0.0 for p in pred]
            if metric == "ndcgAt":
                dcg = sum(h / math.log2(i + 2) for i, h in enumerate(hits))
                idcg = sum(1.0 / math.log2(i + 2)
                           for i in range(min(len(truth), k)))
                vals.append(dcg / idcg if idcg else 0.0)
            elif metric == "map":
                num, ap = 0, 0.0
                for i, h in enumerate(hits):
                    if h:
                        num += 1
                        ap += num / (i + 1)
                vals.append(ap / min(len(truth), k))
            elif metric == "precisionAtk":
                vals.append(sum(hits) / k)
            else:  # recallAtK
                vals.append(sum(hits) / len(truth))
        return float(np.mean(vals)) if vals else 0.0

    def _transform(self, df):
        return df


@register
class RankingAdapter(Transformer):
    """Adapt a fitted recommender into (prediction-list, label-list) rows for
    RankingEvaluator (core/.../recommendation/RankingAdapter parity)."""
    recommenderModel = Param("recommenderModel", "fitted SARModel", None,
                             is_complex=True)
    userCol = Param("userCol", "user index column", "userIdx")
    itemCol = Param("itemCol", "item index column", "itemIdx")
    k = Param("k", "recommendations per user", 10, toInt)

    def _transform(self, df: pd.DataFrame) -> pd.DataFrame:
        model = self.get("recommenderModel")
        recs = model.recommendForAllUsers(self.get("k"), remove_seen=False)
        ucol, icol = self.get("userCol"), self.get("itemCol")
        truth = df.groupby(ucol)[icol].apply(list)
        rec_map = {int(r[ucol]): [x[icol] for x in r["recommendations"]]
                   for _, r in recs.iterrows()}
        rows = []
        for u, items in truth.items():
            rows.append({ucol: u, "prediction": rec_map.get(int(u), []),
                         "label": items})
        return pd.DataFrame(rows)


@register
class RankingTrainValidationSplit(Estimator):
    """Per-user temporal/random split + fit + ranking evaluation
    (core/.../recommendation/RankingTrainValidationSplit parity, incl. the
    min-ratings filtering that avoids user/item cold start)."""
    estimator = Param("estimator", "recommender estimator (e.g. SAR)", None,
                      is_complex=True)
    userCol = Param("userCol", "user index column", "userIdx")
    itemCol = Param("itemCol", "item index column", "itemIdx")
    ratingCol = Param("ratingCol", "rating column", "rating")
    trainRatio = Param("trainRatio", "train fraction per user", 0.75, toFloat)
    minRatingsPerUser = Param("minRatingsPerUser", "drop colder users", 1, toInt)
    minRatingsPerItem = Param("minRatingsPerItem", "drop colder items", 1, toInt)
    k = Param("k", "eval cutoff", 10, toInt)
    seed = Param("seed", "split seed", 0, toInt)

    def _fit(self, df: pd.DataFrame):
        rng = np.random.default_rng(self.get("seed"))
        ucol, icol = self.get("userCol"), self.get("itemCol")
        # cold-start filtering
        uc = df[ucol].value_counts()
        ic = df[icol].value_counts()
        df = df[df[ucol].isin(uc[uc >= self.get("minRatingsPerUser")].index)
                & df[icol].isin(ic[ic >= self.get("minRatingsPerItem")].index)]
        train_idx = []
        val_idx = []
        for _, g in df.groupby(ucol, sort=False):
            idx = g.index.to_numpy()
            rng.shuffle(idx)
            cut = max(1, int(len(idx) * self.get("trainRatio")))
            train_idx.extend(idx[:cut])
            val_idx.extend(idx[cut:])
        train = df.loc[train_idx]
        valid = df.loc[val_idx]
        model = self.get("estimator").fit(train)
        adapter = RankingAdapter(recommenderModel=model, userCol=ucol,
                                 itemCol=icol, k=self.get("k"))
        ranked = adapter.transform(valid) if len(valid) else pd.DataFrame(
            columns=[ucol, "prediction", "label"])
        ev = RankingEvaluator(k=self.get("k"), metricName="ndcgAt")
        out = RankingTrainValidationSplitModel(best=model)
        out.set("validationMetric",
                ev.evaluate(ranked) if len(ranked) else 0.0)
        return out


@register
class RankingTrainValidationSplitModel(Model):
    bestModel = Param("bestModel", "fitted recommender", None, is_complex=True)
    validationMetric = Param("validationMetric", "NDCG@k on validation", None)

    def __init__(self, best=None, **kwargs):
        super().__init__(**kwargs)
        if best is not None:
            self.set("bestModel", best)

    def _transform(self, df):
        return self.get("bestModel").transform(df)
