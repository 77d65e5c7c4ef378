

def test_text_featurizer_stop_words_and_binary():
    import numpy as np
    import pandas as pd
    from mmlspark_amd.stages.text import TextFeaturizer
    df = pd.DataFrame({"text": ["the cat and the hat", "cat cat cat"]})
    base = TextFeaturizer(useIDF=False).fit(df).transform(df)
    nosw = TextFeaturizer(useIDF=False, useStopWordsRemover=True).fit(df) \
        .transform(df)
    # stop words removed → fewer nonzero slots in row 0
    assert len(nosw["features"].iloc[0].indices) < \
        len(base["features"].iloc[0].indices)
    binf = TextFeaturizer(useIDF=False, binary=True).fit(df).transform(df)
    assert set(np.asarray(binf["features"].iloc[1].values).tolist()) == {1.0}
    # custom stop list, case-sensitive
    cs = TextFeaturizer(useIDF=False, useStopWordsRemover=True,
                        stopWords="Cat", caseSensitiveStopWords=True,
                        toLowercase=False).fit(df).transform(df)
    assert len(cs["features"].iloc[1].indices) == 1  # 'cat' kept ('Cat' listed)


def test_train_classifier_reindex_label_off():
    import numpy as np
    import pandas as pd
    from mmlspark_amd.stages.train import TrainClassifier
    from mmlspark_amd.models.gbdt.estimators import LightGBMClassifier
    rng = np.random.default_rng(0)
    X = rng.normal(size=(300, 4)).astype(np.float32)
    y = (X[:, 0] > 0).astype(np.float64)
    df = pd.DataFrame({"f0": X[:, 0], "f1": X[:, 1], "f2": X[:, 2],
                       "f3": X[:, 3], "label": y})
    m = TrainClassifier(model=LightGBMClassifier(numIterations=5, numLeaves=7),
                        reindexLabel=False).fit(df)
    out = m.transform(df)
    acc = (out["prediction"].astype(float).to_numpy() == y).mean()
    assert acc > 0.85
