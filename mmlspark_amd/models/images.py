"""ImageTransformer / ImageSetAugmenter — the opencv module re-expressed on
torch image ops (opencv/src/main/scala/.../ImageTransformer.scala:282:
ResizeImage:42, CropImage:75, ColorFormat:102, Flip:122, Blur:148,
Threshold:172, GaussianKernel:199; ImageSetAugmenter.scala:18).

Stages are a recorded op list applied per image; tensors run on whatever
device is default, so the same code drives MIOpen-backed GPU kernels on
MI355X (interpolate/conv2d) — no OpenCV JNI."""
from __future__ import annotations


import numpy as np
import pandas as pd
import torch

from ..core.param import Param, toList, toString, toInt
from ..core.pipeline import Transformer
from ..core.registry import register


def _to_tensor(img) -> torch.Tensor:
    a = np.asarray(img)
    t = torch.from_numpy(np.ascontiguousarray(a)).float()
    if t.ndim == 2:
        t = t.unsqueeze(-1)
    return t  # (H, W, C) float


def _to_array(t: torch.Tensor, as_uint8: bool) -> np.ndarray:
    # clamp only on the uint8 path: float outputs (e.g. normalize, whose
    # values are signed) must pass through untouched
    if as_uint8:
        return t.clamp(0, 255).cpu().numpy().astype(np.uint8)
    return t.cpu().numpy()


def _apply_stage(t: torch.Tensor, stage: dict) -> torch.Tensor:
    op = stage["op"]
    if op == "resize":
        h, w = int(stage["height"]), int(stage["width"])
        x = t.permute(2, 0, 1).unsqueeze(0)
        x = torch.nn.functional.interpolate(x, size=(h, w), mode="bilinear",
                                            align_corners=False)
        return x.squeeze(0).permute(1, 2, 0)
    if op == "crop":
        x, y = int(stage["x"]), int(stage["y"])
        h, w = int(stage["height"]), int(stage["width"])
        return t[y:y + h, x:x + w]
    if op == "flip":
        code = int(stage.get("flipCode", 1))  # 1=horizontal, 0=vertical, -1=both
        if code >= 1:
            return torch.flip(t, dims=[1])
        if code == 0:
            return torch.flip(t, dims=[0])
        return torch.flip(t, dims=[0, 1])
    if op == "colorFormat":
        fmt = stage.get("format", "gray")
        if fmt in ("gray", "grayscale"):
            if t.shape[2] >= 3:
                gray = (0.299 * t[:, :, 2] + 0.587 * t[:, :, 1]
                        + 0.114 * t[:, :, 0])  # BGR convention like OpenCV
                return gray.unsqueeze(-1)
            return t
        if fmt == "bgr2rgb" and t.shape[2] >= 3:
            return t.flip(-1)
        return t
    if op == "blur":
        kh, kw = int(stage["height"]), int(stage["width"])
        kernel = torch.ones(1, 1, kh, kw, dtype=t.dtype) / (kh * kw)
        x = t.permute(2, 0, 1).unsqueeze(1)  # (C,1,H,W)
        x = torch.nn.functional.conv2d(x, kernel,
                                       padding=(kh // 2, kw // 2))
        return x.squeeze(1).permute(1, 2, 0)[:t.shape[0], :t.shape[1]]
    if op == "gaussianKernel":
        size = int(stage.get("apertureSize", 3))
        sigma = float(stage.get("sigma", 1.0))
        ax = torch.arange(size, dtype=t.dtype) - (size - 1) / 2.0
        g1 = torch.exp(-(ax ** 2) / (2 * sigma * sigma))
        k = (g1[:, None] * g1[None, :])
        k = k / k.sum()
        x = t.permute(2, 0, 1).unsqueeze(1)
        x = torch.nn.functional.conv2d(x, k.reshape(1, 1, size, size),
                                       padding=size // 2)
        return x.squeeze(1).permute(1, 2, 0)[:t.shape[0], :t.shape[1]]
    if op == "threshold":
        thr = float(stage["threshold"])
        mx = float(stage.get("maxVal", 255.0))
        kind = stage.get("thresholdType", "binary")
        if kind == "binary":
            return torch.where(t > thr, torch.full_like(t, mx),
                               torch.zeros_like(t))
        if kind == "binary_inv":
            return torch.where(t > thr, torch.zeros_like(t),
                               torch.full_like(t, mx))
        if kind == "trunc":
            return t.clamp_max(thr)
        if kind == "tozero":
            return torch.where(t > thr, t, torch.zeros_like(t))
        return torch.where(t > thr, torch.zeros_like(t), t)  # tozero_inv
    if op == "normalize":
        mean = torch.tensor(stage.get("mean", [0.0]), dtype=t.dtype)
        std = torch.tensor(stage.get("std", [1.0]), dtype=t.dtype)
        return (t / 255.0 - mean) / std
    raise ValueError(f"unknown image op {op!r}")


def _apply_stage_batched(t: torch.Tensor, stage: dict) -> torch.Tensor:
    """Batched (N, H, W, C) mirror of _apply_stage — uniform-shape batches
    run every op as ONE device launch instead of a per-image Python loop."""
    op = stage["op"]
    if op == "resize":
        h, w = int(stage["height"]), int(stage["width"])
        x = t.permute(0, 3, 1, 2)
        x = torch.nn.functional.interpolate(x, size=(h, w), mode="bilinear",
                                            align_corners=False)
        return x.permute(0, 2, 3, 1)
    if op == "crop":
        x, y = int(stage["x"]), int(stage["y"])
        h, w = int(stage["height"]), int(stage["width"])
        return t[:, y:y + h, x:x + w]
    if op == "flip":
        code = int(stage.get("flipCode", 1))
        if code >= 1:
            return torch.flip(t, dims=[2])
        if code == 0:
            return torch.flip(t, dims=[1])
        return torch.flip(t, dims=[1, 2])
    if op == "colorFormat":
        fmt = stage.get("format", "gray")
        if fmt in ("gray", "grayscale"):
            if t.shape[-1] >= 3:
                gray = (0.299 * t[..., 2] + 0.587 * t[..., 1]
                        + 0.114 * t[..., 0])
                return gray.unsqueeze(-1)
            return t
        if fmt == "bgr2rgb" and t.shape[-1] >= 3:
            return t.flip(-1)
        return t
    if op in ("blur", "gaussianKernel"):
        if op == "blur":
            kh, kw = int(stage["height"]), int(stage["width"])
            k = torch.ones(1, 1, kh, kw, dtype=t.dtype,
                           device=t.device) / (kh * kw)
        else:
            size = int(stage.get("apertureSize", 3))
            sigma = float(stage.get("sigma", 1.0))
            ax = (torch.arange(size, dtype=t.dtype, device=t.device)
                  - (size - 1) / 2.0)
            g1 = torch.exp(-(ax ** 2) / (2 * sigma * sigma))
            k = (g1[:, None] * g1[None, :])
            k = (k / k.sum()).reshape(1, 1, size, size)
            kh = kw = size
        N, H, W, C = t.shape
        x = t.permute(0, 3, 1, 2).reshape(N * C, 1, H, W)
        x = torch.nn.functional.conv2d(x, k, padding=(kh // 2, kw // 2))
        x = x[:, :, :H, :W].reshape(N, C, H, W).permute(0, 2, 3, 1)
        return x
    if op == "threshold":
        return _apply_stage(t, stage)   # purely elementwise
    if op == "normalize":
        mean = torch.tensor(stage.get("mean", [0.0]), dtype=t.dtype,
                            device=t.device)
        std = torch.tensor(stage.get("std", [1.0]), dtype=t.dtype,
                           device=t.device)
        return (t / 255.0 - mean) / std
    raise ValueError(f"unknown image op {op!r}")


@register
class ImageTransformer(Transformer):
    """Stage-list image pipeline. Stages added with fluent helpers
    (ImageTransformer.scala:42-225) or via the ``stages`` param."""
    inputCol = Param("inputCol", "image column", "image")
    outputCol = Param("outputCol", "output image column", "out_image")
    stages = Param("stages", "ordered op list", None, toList)
    outputType = Param("outputType", "uint8|float", "uint8", toString)

    def _add(self, **stage):
        stages = list(self.get("stages") or [])
        stages.append(stage)
        self.set("stages", stages)
        return self

    def resize(self, height: int, width: int):
        return self._add(op="resize", height=height, width=width)

    def crop(self, x: int, y: int, height: int, width: int):
        return self._add(op="crop", x=x, y=y, height=height, width=width)

    def flip(self, flipCode: int = 1):
        return self._add(op="flip", flipCode=flipCode)

    def colorFormat(self, format: str):
        return self._add(op="colorFormat", format=format)

    def blur(self, height: int, width: int):
        return self._add(op="blur", height=height, width=width)

    def gaussianKernel(self, apertureSize: int, sigma: float):
        return self._add(op="gaussianKernel", apertureSize=apertureSize,
                         sigma=sigma)

    def threshold(self, threshold: float, maxVal: float = 255.0,
                  thresholdType: str = "binary"):
        return self._add(op="threshold", threshold=threshold, maxVal=maxVal,
                         thresholdType=thresholdType)

    def normalize(self, mean, std):
        return self._add(op="normalize", mean=mean, std=std)

    def _transform(self, df: pd.DataFrame) -> pd.DataFrame:
        from ..utils.devices import default_device
        stages = self.get("stages") or []
        keep_u8 = (self.get("outputType") == "uint8"
                   and not any(s["op"] == "normalize" for s in stages))
        vals = df[self.get("inputCol")].to_numpy()
        outs = []
        # uniform-shape batches run each op as one device launch
        uniform = (len(vals) > 1 and all(
            isinstance(v, np.ndarray) and v.shape == vals[0].shape
            for v in vals))
        if uniform:
            device = default_device("auto")
            chunk = max(1, int((1 << 28) / max(1, vals[0].size * 4)))
            for s0 in range(0, len(vals), chunk):
                t = torch.from_numpy(
                    np.stack(vals[s0:s0 + chunk])).float().to(device)
                if t.ndim == 3:
                    t = t.unsqueeze(-1)
                for st in stages:
                    t = _apply_stage_batched(t, st)
                if keep_u8:
                    a = t.clamp(0, 255).cpu().numpy().astype(np.uint8)
                else:
                    a = t.cpu().numpy()
                outs.extend(list(a))
        else:
            for img in vals:
                t = _to_tensor(img)
                for st in stages:
                    t = _apply_stage(t, st)
                outs.append(_to_array(t, keep_u8))
        out = df.copy()
        out[self.get("outputCol")] = outs
        return out


@register
class ImageSetAugmenter(Transformer):
    """Dataset augmentation by flips (ImageSetAugmenter.scala:18): emits the
    original rows plus flipped copies."""
    inputCol = Param("inputCol", "image column", "image")
    outputCol = Param("outputCol", "output column", "image")
    flipLeftRight = Param("flipLeftRight", "add LR flips", True)
    flipUpDown = Param("flipUpDown", "add UD flips", False)

    def _transform(self, df: pd.DataFrame) -> pd.DataFrame:
        frames = []
        base = df.copy()
        base[self.get("outputCol")] = df[self.get("inputCol")]
        frames.append(base)
        if self.get("flipLeftRight"):
            f = df.copy()
            f[self.get("outputCol")] = [np.ascontiguousarray(np.asarray(v)[:, ::-1])
                                        for v in df[self.get("inputCol")]]
            frames.append(f)
        if self.get("flipUpDown"):
            f = df.copy()
            f[self.get("outputCol")] = [np.ascontiguousarray(np.asarray(v)[::-1])
                                        for v in df[self.get("inputCol")]]
            frames.append(f)
        return pd.concat(frames, ignore_index=True)


@register
class ResizeImageTransformer(Transformer):
    """Standalone resize (image/ResizeImageTransformer.scala): bilinear
    resize to (height, width), preserving dtype."""
    inputCol = Param("inputCol", "image column", "image")
    outputCol = Param("outputCol", "output column", "image")
    height = Param("height", "target height", 224, toInt)
    width = Param("width", "target width", 224, toInt)

    def _transform(self, df: pd.DataFrame) -> pd.DataFrame:
        h, w = self.get("height"), self.get("width")
        out = df.copy()
        out[self.get("outputCol")] = [
            _to_array(_apply_stage(_to_tensor(np.asarray(v)),
                                   {"op": "resize", "height": h, "width": w}),
                      np.asarray(v).dtype == np.uint8)
            for v in df[self.get("inputCol")]]
        return out


@register
class UnrollImage(Transformer):
    """Image → flat float vector in CHANNEL-MAJOR (CHW) order with raw
    0-255 values — exactly the reference's rearrangement
    (image/UnrollImage.scala:30-55: index = h*W*C + w*C + c, emitted
    channel-by-channel)."""
    inputCol = Param("inputCol", "image column", "image")
    outputCol = Param("outputCol", "unrolled vector column", "unrolled")

    def _transform(self, df: pd.DataFrame) -> pd.DataFrame:
        out = df.copy()
        vecs = []
        for v in df[self.get("inputCol")]:
            a = np.asarray(v)
            if a.ndim == 2:
                a = a[:, :, None]
            vecs.append(np.ascontiguousarray(
                a.transpose(2, 0, 1)).reshape(-1).astype(np.float64))
        out[self.get("outputCol")] = vecs
        return out


@register
class UnrollBinaryImage(Transformer):
    """Encoded image bytes → decoded (+optional resize) → flat vector
    (image/UnrollImage.scala:186 UnrollBinaryImage)."""
    inputCol = Param("inputCol", "image-bytes column", "data")
    outputCol = Param("outputCol", "unrolled vector column", "unrolled")
    height = Param("height", "optional resize height (0 = keep)", 0, toInt)
    width = Param("width", "optional resize width (0 = keep)", 0, toInt)

    def _transform(self, df: pd.DataFrame) -> pd.DataFrame:
        from ..io_http.files import decode_image
        h, w = self.get("height"), self.get("width")
        out = df.copy()
        vecs = []
        for b in df[self.get("inputCol")]:
            img = decode_image(bytes(b))
            if h and w:
                img = _to_array(_apply_stage(
                    _to_tensor(img), {"op": "resize", "height": h,
                                      "width": w}), True)
            if img.ndim == 2:
                img = img[:, :, None]
            vecs.append(np.ascontiguousarray(
                img.transpose(2, 0, 1)).reshape(-1).astype(np.float64))
        out[self.get("outputCol")] = vecs
        return out
