"""Stages: basic transformers, batching, featurize, text, train, automl."""
import numpy as np
import pandas as pd
import pytest

from mmlspark_amd.stages.automl import (DiscreteHyperParam, FindBestModel,
                                        HyperparamBuilder, RangeHyperParam,
                                        TuneHyperparameters)
from mmlspark_amd.stages.basic import (Cacher, DropColumns, EnsembleByKey,
                                       Explode, Lambda, MultiColumnAdapter,
                                       RenameColumn, SelectColumns,
                                       SummarizeData, TextPreprocessor, Timer,
                                       UDFTransformer, UnicodeNormalize)
from mmlspark_amd.stages.batching import (DynamicMiniBatchTransformer,
                                          FixedMiniBatchTransformer,
                                          FlattenBatch, PartitionConsolidator)
from mmlspark_amd.stages.featurize import (CleanMissingData, CountSelector,
                                           DataConversion, Featurize,
                                           ValueIndexer)
from mmlspark_amd.stages.text import MultiNGram, PageSplitter, TextFeaturizer
from mmlspark_amd.stages.train import (ComputeModelStatistics,
                                       ComputePerInstanceStatistics,
                                       TrainClassifier, TrainRegressor)


def test_basic_column_stages():
    df = pd.DataFrame({"a": [1, 2], "b": [3, 4], "c": [5, 6]})
    assert list(DropColumns(cols=["b"]).transform(df).columns) == ["a", "c"]
    assert list(SelectColumns(cols=["b"]).transform(df).columns) == ["b"]
    assert "z" in RenameColumn(inputCol="a", outputCol="z").transform(df).columns
    out = UDFTransformer(udf=lambda x: x * 10, inputCol="a",
                         outputCol="a10").transform(df)
    assert out["a10"].tolist() == [10, 20]
    out = Lambda(fn=lambda d: d.head(1)).transform(df)
    assert len(out) == 1
    assert len(Cacher().transform(df)) == 2


def test_explode_and_ensemble():
    df = pd.DataFrame({"k": ["x", "x", "y"], "v": [1.0, 3.0, 5.0],
                       "lst": [[1, 2], [3], [4]]})
    ex = Explode(inputCol="lst", outputCol="e").transform(df)
    assert len(ex) == 4
    ens = EnsembleByKey(keys=["k"], cols=["v"]).transform(df)
    assert len(ens) == 2
    assert ens[ens["k"] == "x"]["v_ensemble"].iloc[0] == 2.0


def test_minibatch_roundtrip():
    df = pd.DataFrame({"a": list(range(10)), "b": [f"s{i}" for i in range(10)]})
    batched = FixedMiniBatchTransformer(batchSize=3).transform(df)
    assert len(batched) == 4
    assert batched["a"].iloc[0] == [0, 1, 2]
    flat = FlattenBatch().transform(batched)
    assert flat["a"].tolist() == list(range(10))
    assert flat["b"].tolist() == [f"s{i}" for i in range(10)]
    b2 = DynamicMiniBatchTransformer().transform(df)
    assert len(b2) == 1


def test_partition_consolidator():
    df = pd.DataFrame({"a": range(20)})
    out = PartitionConsolidator().transform(df)
    assert sorted(out["a"].tolist()) == list(range(20))


def test_summarize_data():
    df = pd.DataFrame({"x": [1.0, 2.0, np.nan], "s": ["a", "b", "b"]})
    out = SummarizeData().transform(df)
    assert len(out) == 2
    row = out[out["Feature"] == "x"].iloc[0]
    assert row["Missing Value Count"] == 1


def test_text_stages():
    df = pd.DataFrame({"text": ["Hello WORLD", "héllo"]})
    out = TextPreprocessor(inputCol="text", outputCol="t",
                           map={"hello": "hi"}).transform(df)
    assert out["t"].iloc[0] == "hi world"
    out = UnicodeNormalize(inputCol="text", outputCol="u").transform(df)
    assert "hello" in out["u"].iloc[1].replace("́", "")
    df2 = pd.DataFrame({"tokens": [["a", "b", "c"]]})
    out = MultiNGram(inputCol="tokens", outputCol="ng",
                     lengths=[1, 2]).transform(df2)
    assert "a b" in out["ng"].iloc[0]
    df3 = pd.DataFrame({"text": ["abcdefghij" * 100]})
    out = PageSplitter(inputCol="text", outputCol="p", maximumPageLength=300,
                       minimumPageLength=100).transform(df3)
    assert all(len(p) <= 300 for p in out["p"].iloc[0])
    assert "".join(out["p"].iloc[0]) == df3["text"].iloc[0]


def test_text_featurizer():
    df = pd.DataFrame({"text": ["the cat sat", "the dog ran fast",
                                "cat and dog"]})
    m = TextFeaturizer(numFeatures=1 << 12, useNGram=True).fit(df)
    out = m.transform(df)
    v = out["features"].iloc[0]
    assert len(v.indices) > 0


def test_featurize_mixed_types():
    df = pd.DataFrame({"num": [1.0, np.nan, 3.0],
                       "cat": ["a", "b", "a"],
                       "label": [0, 1, 0]})
    m = Featurize(inputCols=["num", "cat"]).fit(df)
    out = m.transform(df)
    v = np.stack([np.asarray(x) for x in out["features"]])
    assert v.shape == (3, 3)  # 1 numeric + 2 one-hot
    assert v[1, 0] == 2.0  # mean-imputed


def test_clean_missing_and_indexer_and_conversion():
    df = pd.DataFrame({"x": [1.0, np.nan, 3.0], "c": ["b", "a", "b"]})
    m = CleanMissingData(inputCols=["x"], cleaningMode="Median").fit(df)
    assert m.transform(df)["x"].iloc[1] == 2.0
    vi = ValueIndexer(inputCol="c").fit(df)
    out = vi.transform(df)
    assert out["c_idx"].tolist() == [1, 0, 1]
    dc = DataConversion(cols=["x"], convertTo="string").transform(df)
    assert isinstance(dc["x"].iloc[0], str)


def test_count_selector():
    df = pd.DataFrame({"features": [np.array([1.0, 0.0, 2.0]),
                                    np.array([0.0, 0.0, 1.0])]})
    m = CountSelector().fit(df)
    out = m.transform(df)
    assert len(out["features"].iloc[0]) == 2


@pytest.fixture(scope="module")
def mixed_class_df():
    rng = np.random.default_rng(0)
    n = 800
    return pd.DataFrame({
        "age": rng.integers(18, 80, n).astype(float),
        "city": rng.choice(["nyc", "sf", "chi"], n),
        "income": rng.normal(50, 10, n),
        "label": rng.integers(0, 2, n),
    }).assign(label=lambda d: ((d.age > 40) & (d.income > 50)).astype(int))


def test_train_classifier_auto_featurize(mixed_class_df):
    m = TrainClassifier().fit(mixed_class_df)
    out = m.transform(mixed_class_df)
    acc = (out["prediction"].to_numpy() ==
           mixed_class_df["label"].to_numpy()).mean()
    assert acc > 0.9
    stats = ComputeModelStatistics(evaluationMetric="classification").transform(out)
    assert stats["accuracy"].iloc[0] > 0.9
    assert "AUC" in stats.columns
    per = ComputePerInstanceStatistics().transform(out)
    assert "log_loss" in per.columns


def test_train_regressor():
    rng = np.random.default_rng(1)
    df = pd.DataFrame({"x1": rng.normal(size=500),
                       "x2": rng.normal(size=500)})
    df["label"] = 2 * df.x1 - df.x2 + rng.normal(size=500) * 0.1
    m = TrainRegressor().fit(df)
    out = m.transform(df)
    stats = ComputeModelStatistics(evaluationMetric="regression",
                                   scoredLabelsCol="prediction").transform(out)
    assert stats["R^2"].iloc[0] > 0.8


def test_tune_hyperparameters(mixed_class_df):
    from mmlspark_amd.models.gbdt.estimators import LightGBMClassifier
    from mmlspark_amd.stages.featurize import Featurize
    feats = Featurize(inputCols=["age", "city", "income"]).fit(mixed_class_df)
    dff = feats.transform(mixed_class_df)
    space = (HyperparamBuilder()
             .addHyperparam("numLeaves", DiscreteHyperParam([4, 15]))
             .addHyperparam("numIterations", RangeHyperParam(5, 15, is_int=True))
             .build())
    tuner = TuneHyperparameters(
        models=[LightGBMClassifier(featuresCol="features")],
        paramSpace=space, numRuns=3, numFolds=2, evaluationMetric="accuracy")
    best = tuner.fit(dff)
    out = best.transform(dff)
    assert (out["prediction"].to_numpy() ==
            dff["label"].to_numpy()).mean() > 0.85
    assert best.get("bestMetric") > 0.8


def test_find_best_model(mixed_class_df):
    from mmlspark_amd.models.gbdt.estimators import LightGBMClassifier
    from mmlspark_amd.stages.featurize import Featurize
    feats = Featurize(inputCols=["age", "city", "income"]).fit(mixed_class_df)
    dff = feats.transform(mixed_class_df)
    m1 = LightGBMClassifier(numIterations=2, numLeaves=2).fit(dff)
    m2 = LightGBMClassifier(numIterations=20, numLeaves=15).fit(dff)
    fb = FindBestModel(models=[m1, m2], evaluationMetric="accuracy").fit(dff)
    assert fb.get("bestModel").uid == m2.uid
    assert len(fb.getEvaluationResults()) == 2


def test_class_balancer():
    from mmlspark_amd.stages.basic import ClassBalancer
    df = pd.DataFrame({"label": [0, 0, 0, 1]})
    m = ClassBalancer(inputCol="label").fit(df)
    out = m.transform(df)
    assert out["weight"].tolist() == [1.0, 1.0, 1.0, 3.0]


def test_stratified_repartition():
    from mmlspark_amd.stages.basic import StratifiedRepartition
    df = pd.DataFrame({"label": [0] * 8 + [1] * 8, "x": range(16)})
    out = StratifiedRepartition(labelCol="label").transform(df)
    # every contiguous half (2-rank shard) sees both classes evenly
    first, second = out.head(8), out.tail(8)
    assert first["label"].sum() == 4 and second["label"].sum() == 4


def test_udfs_and_fluent(binary_df):
    from mmlspark_amd.stages.basic import DropColumns
    from mmlspark_amd.stages.udfs import (get_value_at, ml_transform,
                                          to_vector, vector_to_array)
    df = pd.DataFrame({"a": [1.0, 2.0], "b": [3.0, 4.0]})
    v = to_vector(df, ["a", "b"])
    assert list(v["features"].iloc[0]) == [1.0, 3.0]
    g = get_value_at(v, "features", 1, "b_again")
    assert g["b_again"].tolist() == [3.0, 4.0]
    arr = vector_to_array(v, "features", "arr")
    assert arr["arr"].iloc[1] == [2.0, 4.0]
    out = ml_transform(df, DropColumns(cols=["b"]))
    assert list(out.columns) == ["a"]


def test_model_downloader(tmp_path):
    import torch
    from mmlspark_amd.models.downloader import ModelDownloader
    repo = ModelDownloader(str(tmp_path))
    lin = torch.nn.Linear(3, 2)
    schema = repo.publish("lin", lin, dataset="synthetic")
    assert schema.size > 0
    got = repo.download_by_name("lin")
    state = repo.load_state("lin")
    assert "weight" in state
    # tamper → verification fails
    with open(got.uri, "ab") as f:
        f.write(b"x")
    import pytest as _pt
    with _pt.raises(IOError):
        repo.download_by_name("lin")


def test_tune_hyperparameters_grid(mixed_class_df):
    from mmlspark_amd.models.gbdt.estimators import LightGBMClassifier
    from mmlspark_amd.stages.featurize import Featurize
    feats = Featurize(inputCols=["age", "city", "income"]).fit(mixed_class_df)
    dff = feats.transform(mixed_class_df)
    space = (HyperparamBuilder()
             .addHyperparam("numLeaves", DiscreteHyperParam([4, 8]))
             .addHyperparam("numIterations", DiscreteHyperParam([5, 10]))
             .build())
    tuner = TuneHyperparameters(
        models=[LightGBMClassifier(featuresCol="features")],
        paramSpace=space, searchMode="grid", numFolds=2,
        evaluationMetric="accuracy")
    best = tuner.fit(dff)
    assert best.get("bestMetric") > 0.8
    assert best.get("bestParams")["numLeaves"] in (4, 8)


def test_plot_helpers(tmp_path):
    """plot/ module (plot.py:17,45 analog): AUC math vs sklearn oracle and
    confusion-matrix counts; renders to an Agg canvas (no display)."""
    import matplotlib
    matplotlib.use("Agg")
    import matplotlib.pyplot as plt
    import pandas as pd
    from sklearn.metrics import roc_auc_score
    from mmlspark_amd.plot import confusionMatrix, roc

    rng = np.random.default_rng(0)
    y = rng.integers(0, 2, 500)
    score = y * 0.6 + rng.random(500) * 0.7
    df = pd.DataFrame({"label": y.astype(float), "score": score,
                       "pred": (score > 0.65).astype(float)})
    plt.figure()
    auc = roc(df, "label", "score")
    plt.savefig(tmp_path / "roc.png")
    assert abs(auc - roc_auc_score(y, score)) < 1e-9
    plt.figure()
    cm = confusionMatrix(df, "label", "pred", ["neg", "pos"])
    plt.savefig(tmp_path / "cm.png")
    assert cm.sum() == 500
    assert cm[1, 1] > cm[1, 0]  # classifier is informative
    plt.close("all")


def test_fluent_api_monkeypatch(binary_df):
    """install_fluent_api (FluentAPI.py parity): df.mlFit / df.mlTransform."""
    from mmlspark_amd.stages.basic import SelectColumns
    from mmlspark_amd.stages.udfs import install_fluent_api
    from mmlspark_amd.models.gbdt.estimators import LightGBMClassifier
    install_fluent_api()
    model = binary_df.mlFit(LightGBMClassifier(numIterations=5, numLeaves=7))
    out = binary_df.mlTransform(
        model, SelectColumns(cols=["prediction", "label"]))
    assert list(out.columns) == ["prediction", "label"]
    acc = (out["prediction"] == out["label"]).mean()
    assert acc > 0.8


def test_fast_vector_assembler():
    from mmlspark_amd.stages.featurize import FastVectorAssembler
    from mmlspark_amd.core.schema import SparseVector
    df = pd.DataFrame({
        "a": [1.0, 2.0, 3.0],
        "v": [np.array([1, 2], dtype=np.float32)] * 3,
        "s": [SparseVector(4, [1], [5.0]), SparseVector(4, [0, 3], [1.0, 2.0]),
              SparseVector(4, [], [])],
    })
    out = FastVectorAssembler(inputCols=["a", "v", "s"], outputCol="f") \
        .transform(df)
    f = np.stack(out["f"].to_numpy())
    assert f.shape == (3, 7)
    np.testing.assert_allclose(f[0], [1, 1, 2, 0, 5, 0, 0])
    np.testing.assert_allclose(f[1], [2, 1, 2, 1, 0, 0, 2])
    np.testing.assert_allclose(f[2], [3, 1, 2, 0, 0, 0, 0])


def test_shared_variable_and_singleton():
    from mmlspark_amd.utils.shared import (SharedSingleton, SharedVariable,
                                           clear_pool)
    clear_pool()
    calls = []

    def make():
        calls.append(1)
        return object()

    sv = SharedVariable(make)
    assert sv.get() is sv.get()
    assert len(calls) == 1
    # singleton: two instances from the same ctor share one value
    s1, s2 = SharedSingleton(make), SharedSingleton(make)
    assert s1.get() is s2.get()
    # pickling carries the key, not the value (module-level ctor so the
    # thunk itself pickles)
    import pickle
    svp = SharedVariable(dict)
    first = svp.get()
    sv2 = pickle.loads(pickle.dumps(svp))
    assert sv2.get() is first


def test_cluster_topology(monkeypatch):
    from mmlspark_amd.parallel.cluster import (get_driver_host,
                                               get_num_executors,
                                               get_topology)
    monkeypatch.setenv("WORLD_SIZE", "8")
    monkeypatch.setenv("RANK", "3")
    monkeypatch.setenv("LOCAL_RANK", "3")
    monkeypatch.setenv("LOCAL_WORLD_SIZE", "4")
    monkeypatch.setenv("MASTER_ADDR", "10.0.0.1")
    t = get_topology()
    assert (t.world_size, t.rank, t.local_rank) == (8, 3, 3)
    assert t.n_nodes == 2
    assert get_num_executors() == 8
    assert get_driver_host() == "10.0.0.1"
