"""TorchModel + ImageFeaturizer — the deep-learning module on PyTorch-ROCm.

Parity targets:
  * TorchModel ≈ CNTKModel (deep-learning/.../cntk/CNTKModel.scala:35-140):
    batched DNN inference on a broadcast network — minibatch rows → tensors →
    forward → unbatch to an output vector column.
  * ImageFeaturizer (cntk/ImageFeaturizer.scala:41): resize → normalize →
    backbone truncated at ``cutOutputLayers``; transfer learning head +
    fine-tune support.  DP=8 runs the backbone under DDP over RCCL.
No model zoo is reachable offline, so weights are random-init unless a
``modelPath`` state-dict is given (the ModelDownloader analog is a local
state-dict repo).
"""
from __future__ import annotations

from typing import Optional

import numpy as np
import pandas as pd
import torch

from ..core.param import Param, toBool, toFloat, toInt, toString
from ..core.pipeline import Estimator, Model
from ..core.registry import register
from ..core.schema import matrix_to_vector_column
from ..utils.devices import default_device
from .resnet import MODELS


def _rows_to_batch(vals, size: int) -> torch.Tensor:
    """Image cells (H,W,C uint8 | C,H,W float | flat | encoded bytes)
    -> (N,3,size,size) f32.  Encoded bytes (JPEG/PNG/BMP/PPM) decode
    through io_http.files (native C++ JPEG decoder when built) — the
    bytes-to-features path of the reference's ImageFeaturizer
    (core/.../core/image/ImageUtils.scala)."""
    decoded = None
    if len(vals) and isinstance(vals[0], (bytes, bytearray, memoryview)):
        # decode in a thread pool: the native JPEG decoder releases the GIL
        from concurrent.futures import ThreadPoolExecutor
        from ..io_http.files import decode_image
        with ThreadPoolExecutor(max_workers=min(16, max(1, len(vals)))) as ex:
            decoded = list(ex.map(lambda b: decode_image(bytes(b)), vals))
        if all(d.ndim == 3 and d.shape == decoded[0].shape
               and d.dtype == np.uint8 for d in decoded):
            # uniform decoded batch: one stack + one vectorized normalize
            batch = torch.from_numpy(np.stack(decoded)).permute(
                0, 3, 1, 2).float().div_(255.0)
            if batch.shape[-1] != size or batch.shape[-2] != size:
                batch = torch.nn.functional.interpolate(
                    batch, size=(size, size), mode="bilinear",
                    align_corners=False)
            return batch
    outs = []
    for i, v in enumerate(vals):
        if decoded is not None:
            a = decoded[i]
        elif isinstance(v, (bytes, bytearray, memoryview)):
            from ..io_http.files import decode_image
            a = decode_image(bytes(v))
        else:
            a = np.asarray(v)
        if a.ndim == 1:  # flattened
            side = int(round((a.size / 3) ** 0.5))
            a = a.reshape(3, side, side) if a.size == 3 * side * side else a
        t = torch.from_numpy(np.ascontiguousarray(a)).float()
        if t.ndim == 3 and t.shape[-1] in (1, 3):  # HWC → CHW
            t = t.permute(2, 0, 1)
        if t.ndim == 2:
            t = t.unsqueeze(0).repeat(3, 1, 1)
        if t.shape[0] == 1:
            t = t.repeat(3, 1, 1)
        if t.max() > 1.5:
            t = t / 255.0
        outs.append(t)
    batch = torch.stack(outs)
    if batch.shape[-1] != size or batch.shape[-2] != size:
        batch = torch.nn.functional.interpolate(
            batch, size=(size, size), mode="bilinear", align_corners=False)
    return batch


@register
class TorchModel(Model):
    """Generic batched torch-module inference transformer (CNTKModel analog).

    The module is held once per process (broadcast analog), rows are
    minibatched, coerced to tensors, run under no_grad, and unbatched into an
    output vector column (CNTKModelUtils.applyModel, CNTKModel.scala:89-140).
    """

    inputCol = Param("inputCol", "input column (vector/image)", "input")
    outputCol = Param("outputCol", "output vector column", "output")
    batchSize = Param("batchSize", "inference minibatch size", 64, toInt)
    moduleBytes = Param("moduleBytes", "torchscript/state payload", None,
                        is_complex=True)
    device = Param("device", "cpu|cuda|auto", "auto", toString)

    def __init__(self, module: Optional[torch.nn.Module] = None, **kwargs):
        super().__init__(**kwargs)
        self._module = module
        if module is not None:
            import io
            buf = io.BytesIO()
            torch.save(module, buf)
            self.set("moduleBytes", buf.getvalue())

    def _post_deserialize_init(self):
        b = self.get("moduleBytes")
        if b is not None:
            import io
            self._module = torch.load(io.BytesIO(bytes(b)),
                                      map_location="cpu", weights_only=False)

    @property
    def module(self) -> torch.nn.Module:
        if getattr(self, "_module", None) is None:
            self._post_deserialize_init()
        return self._module

    def _forward(self, batch: torch.Tensor) -> torch.Tensor:
        return self.module(batch)

    def _coerce(self, vals) -> torch.Tensor:
        first = np.asarray(vals[0])
        if first.ndim >= 2:  # images
            return _rows_to_batch(vals, max(first.shape[:2]))
        return torch.from_numpy(np.stack([np.asarray(v, dtype=np.float32)
                                          for v in vals]))

    def _transform(self, df: pd.DataFrame) -> pd.DataFrame:
        device = default_device(self.get("device"))
        mod = self.module.to(device).eval()
        vals = df[self.get("inputCol")].to_numpy()
        bs = self.get("batchSize")
        outs = []
        with torch.no_grad():
            for s in range(0, len(vals), bs):
                batch = self._coerce(vals[s:s + bs]).to(device)
                out = self._forward(batch)
                outs.append(out.detach().float().cpu().numpy())
        mat = np.concatenate(outs) if outs else np.zeros((0, 0), np.float32)
        res = df.copy()
        res[self.get("outputCol")] = matrix_to_vector_column(
            mat.reshape(len(df), -1) if len(df) else mat)
        return res


@register
class ImageFeaturizer(TorchModel):
    """Transfer-learning image featurizer (ImageFeaturizer.scala:41)."""

    modelName = Param("modelName", "backbone: ResNet18/34/50/101", "ResNet50")
    batchSize = Param("batchSize", "inference minibatch size (larger than "
                      "the TorchModel default: decode threads + the conv "
                      "stack both want deep batches)", 256, toInt)
    cutOutputLayers = Param("cutOutputLayers",
                            "how many output layers to cut (1 = pooled "
                            "features)", 1, toInt)
    imageSize = Param("imageSize", "square input resolution", 224, toInt)
    modelPath = Param("modelPath", "optional state-dict path", None)
    inputCol = Param("inputCol", "image column", "image")
    outputCol = Param("outputCol", "feature vector column", "features")

    def __init__(self, **kwargs):
        super().__init__(**kwargs)

    @property
    def module(self) -> torch.nn.Module:
        if getattr(self, "_module", None) is None:
            net = MODELS[self.get("modelName")]()
            path = self.get("modelPath")
            if path:
                net.load_state_dict(torch.load(path, map_location="cpu"))
            self._module = net
        return self._module

    def setModel(self, module: torch.nn.Module):
        self._module = module
        return self

    def _forward(self, batch):
        return self.module(batch, cut_output_layers=self.get("cutOutputLayers"))

    def _coerce(self, vals):
        return _rows_to_batch(vals, self.get("imageSize"))


@register
class DeepVisionClassifier(Estimator):
    """Transfer-learning trainer: backbone + linear head fine-tuned with
    cross-entropy under DDP over RCCL when launched one-process-per-GPU
    (the DP=8 path of BASELINE config #4)."""

    labelCol = Param("labelCol", "label column", "label")
    imageCol = Param("imageCol", "image column", "image")
    predictionCol = Param("predictionCol", "prediction column", "prediction")
    modelName = Param("modelName", "backbone name", "ResNet50")
    imageSize = Param("imageSize", "square input resolution", 224, toInt)
    batchSize = Param("batchSize", "train batch size", 64, toInt)
    epochs = Param("epochs", "training epochs", 1, toInt)
    learningRate = Param("learningRate", "adam lr", 1e-3, toFloat)
    freezeBackbone = Param("freezeBackbone", "train only the head", False, toBool)
    device = Param("device", "cpu|cuda|auto", "auto", toString)

    def _fit(self, df: pd.DataFrame):
        from ..parallel.comm import get_comm
        comm = get_comm()
        device = default_device(self.get("device"))
        y = df[self.get("labelCol")].to_numpy()
        n_classes = int(y.max()) + 1
        net = MODELS[self.get("modelName")](num_classes=n_classes).to(device)
        if self.get("freezeBackbone"):
            for name, p in net.named_parameters():
                if not name.startswith("fc."):
                    p.requires_grad_(False)
        train_net = net
        if comm.is_distributed:
            train_net = torch.nn.parallel.DistributedDataParallel(
                net, device_ids=[device.index] if device.type == "cuda" else None)
        opt = torch.optim.Adam([p for p in train_net.parameters()
                                if p.requires_grad], lr=self.get("learningRate"))
        vals = df[self.get("imageCol")].to_numpy()
        bs = self.get("batchSize")
        size = self.get("imageSize")
        yt = torch.from_numpy(y.astype(np.int64))
        train_net.train()
        for _ in range(self.get("epochs")):
            perm = torch.randperm(len(vals))
            for s in range(0, len(vals), bs):
                sel = perm[s:s + bs]
                batch = _rows_to_batch(vals[sel.numpy()], size).to(device)
                target = yt[sel].to(device)
                opt.zero_grad(set_to_none=True)
                loss = torch.nn.functional.cross_entropy(train_net(batch), target)
                loss.backward()
                opt.step()
        net.eval()
        model = DeepVisionModel(module=net)
        model.set("inputCol", self.get("imageCol"))
        model.set("outputCol", "logits")
        model.set("predictionCol", self.get("predictionCol"))
        model.set("batchSize", bs)
        return model


@register
class DeepVisionModel(TorchModel):
    predictionCol = Param("predictionCol", "prediction column", "prediction")
    probabilityCol = Param("probabilityCol", "softmax probability column",
                           "probability")

    def _transform(self, df):
        out = super()._transform(df)
        logits = np.stack(out[self.get("outputCol")].to_numpy()) if len(out) \
            else np.zeros((0, 1))
        if len(out):
            z = logits - logits.max(axis=1, keepdims=True)
            e = np.exp(z)
            out[self.get("probabilityCol")] = matrix_to_vector_column(
                e / e.sum(axis=1, keepdims=True))
        else:
            out[self.get("probabilityCol")] = []
        out[self.get("predictionCol")] = logits.argmax(axis=1).astype(np.float64) \
            if len(out) else []
        return out


@register
class CNTKModel(TorchModel):
    """Name-compatible alias of TorchModel for reference users
    (cntk/CNTKModel.scala:520-562): batched DNN inference with feed/fetch
    dict naming.  The MI355X runtime holds torch modules, not CNTK graphs,
    so `feedDict`/`fetchDict` map the single model input/output to DataFrame
    columns (CNTKModel.scala:212-226 param surface); multi-head graphs are
    expressed as torch modules returning one tensor per registered output.
    """

    feedDict = Param("feedDict", "model-input-name -> input column "
                     "({name: col}; single entry sets inputCol)", None)
    fetchDict = Param("fetchDict", "output column -> model-output-name "
                      "({col: name}; single entry sets outputCol)", None)
    convertOutputToDenseVector = Param(
        "convertOutputToDenseVector", "emit output as a dense vector column "
        "(always true here — outputs are vector columns)", True, toBool)
    batchInput = Param("batchInput", "minibatch rows before inference "
                       "(always on — the transform is batched)", True, toBool)
    shapeOutput = Param("shapeOutput", "reshape output to the model's "
                        "declared shape (flat vector here)", False, toBool)

    def _transform(self, df: pd.DataFrame) -> pd.DataFrame:
        fd, fc = self.get("feedDict"), self.get("fetchDict")
        if fd:
            self.set("inputCol", next(iter(fd.values())))
        if fc:
            self.set("outputCol", next(iter(fc.keys())))
        return super()._transform(df)
