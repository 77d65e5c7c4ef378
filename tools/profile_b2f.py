"""Phase breakdown of the JPEG bytes -> ResNet features path."""
import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, pandas as pd, torch
from mmlspark_amd.io_http.jpeg_codec import encode_jpeg, decode_jpeg
from mmlspark_amd.models.image_featurizer import ImageFeaturizer, _rows_to_batch

rng = np.random.default_rng(0)
side = 224
yy, xx = np.mgrid[0:side, 0:side]
base = (np.sin(xx / 9.0) * np.cos(yy / 13.0) * 90 + 128)
blobs = []
for i in range(16):
    img = np.clip(np.stack([np.roll(base, i, 0)] * 3, -1) + rng.normal(0, 8, (side, side, 3)), 0, 255)
    blobs.append(encode_jpeg(img.astype(np.uint8), quality=90))
n = 1024
vals = [blobs[i % 16] for i in range(n)]

# 1. decode only (thread pool like the featurizer)
from concurrent.futures import ThreadPoolExecutor
t0 = time.perf_counter()
with ThreadPoolExecutor(max_workers=16) as ex:
    dec = list(ex.map(lambda b: decode_jpeg(b), vals))
t_dec = time.perf_counter() - t0
print(f"decode {n} imgs: {t_dec*1e3:.0f} ms ({n/t_dec:.0f}/s)")

# 2. coerce (stack+normalize)
t0 = time.perf_counter()
batch = _rows_to_batch(vals[:256], 224)
t_coerce = time.perf_counter() - t0
print(f"_rows_to_batch(256): {t_coerce*1e3:.0f} ms")

# 3. forward only
feat = ImageFeaturizer(modelName="ResNet50", cutOutputLayers=1, imageSize=224)
dev = "cuda" if torch.cuda.is_available() else "cpu"
mod = feat.module.to(dev).eval()
x = batch.to(dev)
with torch.no_grad():
    for _ in range(3):
        mod(x, cut_output_layers=1)
    torch.cuda.synchronize() if dev == "cuda" else None
    t0 = time.perf_counter()
    for _ in range(5):
        mod(x, cut_output_layers=1)
    torch.cuda.synchronize() if dev == "cuda" else None
t_fwd = (time.perf_counter() - t0) / 5
print(f"forward bs=256 fp32: {t_fwd*1e3:.0f} ms ({256/t_fwd:.0f} img/s)")

# 4. full transform with default batch
df = pd.DataFrame({"image": vals})
feat.transform(df.head(64))
t0 = time.perf_counter()
feat.transform(df)
t_all = time.perf_counter() - t0
print(f"full transform {n}: {t_all*1e3:.0f} ms ({n/t_all:.0f}/s)")
