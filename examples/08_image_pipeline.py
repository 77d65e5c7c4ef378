"""Image workflow: codecs → ImageTransformer preprocessing → ResNet
featurization → DeepVisionClassifier fine-tune → ImageLIME explanation.
Runs on CPU or MI355X (PyTorch-ROCm path)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import pandas as pd
import torch

from mmlspark_amd.explainers.lime import ImageLIME
from mmlspark_amd.io_http.files import decode_image, encode_image
from mmlspark_amd.models.image_featurizer import (DeepVisionClassifier,
                                                  ImageFeaturizer)
from mmlspark_amd.models.images import ImageTransformer

rng = np.random.default_rng(0)

# synthetic two-class images: red-ish vs blue-ish squares, stored as PNG
# bytes to exercise the pure-numpy codec round trip
def make_img(cls):
    img = rng.integers(0, 60, size=(40, 40, 3)).astype(np.uint8)
    img[8:32, 8:32, 0 if cls == 0 else 2] = 220
    return img

raw = [(encode_image(make_img(i % 2), "png"), i % 2) for i in range(64)]
df = pd.DataFrame({"png": [r[0] for r in raw], "label": [r[1] for r in raw]})
df["image"] = [decode_image(b) for b in df["png"]]

# opencv-module analog: resize + normalize-ish crop pipeline
it = (ImageTransformer(inputCol="image", outputCol="proc")
      .resize(32, 32).crop(2, 2, 28, 28).resize(32, 32))
df = it.transform(df)

feats = ImageFeaturizer(modelName="ResNet18", imageSize=32,
                        inputCol="proc", outputCol="features").transform(df)
print("featurized:", np.stack(feats["features"].to_numpy()).shape)

torch.manual_seed(0)
clf = DeepVisionClassifier(modelName="ResNet18", imageSize=32, epochs=10,
                           batchSize=16, learningRate=5e-3,
                           freezeBackbone=False,
                           imageCol="proc").fit(df)
scored = clf.transform(df)
acc = float((scored["prediction"].to_numpy() == df["label"].to_numpy()).mean())
print("fine-tune accuracy:", acc)

lime = ImageLIME(model=clf, targetCol="logits", targetClasses=[1],
                 inputCol="proc", cellSize=8, numSamples=128)
exp = lime.transform(df.head(1))
print("LIME superpixel weights:", np.round(exp["explanation"].iloc[0][0], 3))
