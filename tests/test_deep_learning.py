"""ImageFeaturizer / TorchModel / DeepVisionClassifier (CPU-sized)."""
import numpy as np
import pandas as pd
import pytest
import torch

from mmlspark_amd.models.image_featurizer import (DeepVisionClassifier,
                                                  ImageFeaturizer, TorchModel)


def _image_df(n=8, hw=32, seed=0):
    rng = np.random.default_rng(seed)
    imgs = [rng.integers(0, 255, size=(hw, hw, 3), dtype=np.uint8).astype(np.uint8)
            for _ in range(n)]
    return pd.DataFrame({"image": imgs})


def test_torch_model_vector_io():
    lin = torch.nn.Linear(4, 2)
    df = pd.DataFrame({"input": [np.ones(4, dtype=np.float32)] * 5})
    m = TorchModel(module=lin, inputCol="input", outputCol="out", device="cpu")
    out = m.transform(df)
    mat = np.stack(out["out"].to_numpy())
    assert mat.shape == (5, 2)
    ref = lin(torch.ones(5, 4)).detach().numpy()
    np.testing.assert_allclose(mat, ref, rtol=1e-5)


def test_torch_model_save_load(tmp_path):
    import os
    lin = torch.nn.Linear(3, 3)
    m = TorchModel(module=lin, inputCol="input", outputCol="out", device="cpu")
    df = pd.DataFrame({"input": [np.arange(3, dtype=np.float32)] * 2})
    o1 = np.stack(m.transform(df)["out"].to_numpy())
    m.save(os.path.join(tmp_path, "tm"))
    m2 = TorchModel.load(os.path.join(tmp_path, "tm"))
    o2 = np.stack(m2.transform(df)["out"].to_numpy())
    np.testing.assert_allclose(o1, o2, rtol=1e-6)


def test_image_featurizer_resnet18_cpu():
    df = _image_df(4, hw=40)
    f = ImageFeaturizer(modelName="ResNet18", imageSize=64, cutOutputLayers=1,
                        device="cpu", batchSize=2)
    out = f.transform(df)
    feats = np.stack(out["features"].to_numpy())
    assert feats.shape == (4, 512)  # ResNet18 feature dim
    assert np.isfinite(feats).all()


def test_image_featurizer_layer_cut():
    df = _image_df(2, hw=32)
    f0 = ImageFeaturizer(modelName="ResNet18", imageSize=32, cutOutputLayers=0,
                         device="cpu")
    logits = np.stack(f0.transform(df)["features"].to_numpy())
    assert logits.shape[1] == 1000


def test_deep_vision_classifier_learns():
    # trivially separable: red vs blue images
    rng = np.random.default_rng(1)
    imgs, ys = [], []
    for i in range(32):
        img = np.zeros((32, 32, 3), dtype=np.uint8)
        c = i % 2
        img[:, :, 0 if c == 0 else 2] = 200 + rng.integers(0, 55)
        imgs.append(img)
        ys.append(c)
    df = pd.DataFrame({"image": imgs, "label": ys})
    torch.manual_seed(0)
    est = DeepVisionClassifier(modelName="ResNet18", imageSize=32, epochs=10,
                               batchSize=8, learningRate=5e-3, device="cpu")
    model = est.fit(df)
    out = model.transform(df)
    acc = (out["prediction"].to_numpy() == np.array(ys)).mean()
    assert acc > 0.8, acc
