"""Cognitive-services pipeline against a local mock endpoint: sentiment +
anomaly detection + error-column handling — the HTTP-on-DataFrame workflow
(cognitive/.../TextAnalytics.scala, AnomalyDetection.scala) without any
Azure dependency.  Runs anywhere (loopback HTTP)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import pandas as pd

from mmlspark_amd.io_http.cognitive import (DetectLastAnomaly,
                                            SimpleDetectAnomalies,
                                            TextSentiment)
from mmlspark_amd.serving.server import ServingServer

# stand-in service: echoes a sentiment-shaped / anomaly-shaped response
def handler(payloads):
    out = []
    for p in payloads:
        if "documents" in p:
            out.append({"documents": [
                {"id": d["id"],
                 "sentiment": "positive" if "good" in d.get("text", "")
                 else "negative"} for d in p["documents"]], "errors": []})
        elif "series" in p:
            vals = [pt["value"] for pt in p["series"]]
            mean = sum(vals) / max(len(vals), 1)
            out.append({"isAnomaly": abs(vals[-1] - mean) > 2 * (max(vals) - min(vals) + 1e-9) / len(vals),
                        "expectedValue": mean})
        else:
            out.append({"echo": p})
    return out


srv = ServingServer(handler, port=0, mode="continuous").start()
url = f"http://127.0.0.1:{srv.port}/"
try:
    reviews = pd.DataFrame({
        "text": ["good plot good cast", "terrible pacing", "good soundtrack"],
        "key": ["k"] * 3,
    })
    scored = TextSentiment(url=url, subscriptionKeyCol="key",
                           outputCol="sentiment").transform(reviews)
    print(scored[["text", "sentiment"]].to_string(index=False))

    series = pd.DataFrame({
        "group": ["sensor-1"] * 6 + ["sensor-2"] * 6,
        "timestamp": [f"2024-01-{d:02d}T00:00:00Z" for d in range(1, 7)] * 2,
        "value": [1.0, 1.1, 0.9, 1.0, 1.05, 9.5,     # spike at the end
                  5.0, 5.1, 4.9, 5.0, 5.2, 5.1],
    })
    det = SimpleDetectAnomalies(url=url, outputCol="anomaly")
    out = det.transform(series)
    print(out[["group", "value", "anomaly"]].tail(4).to_string(index=False))

    last = DetectLastAnomaly(url=url, seriesCol="series", outputCol="verdict")
    one = pd.DataFrame({"series": [[{"timestamp": t, "value": v}
                                    for t, v in zip(series.timestamp[:6],
                                                    series.value[:6])]]})
    print(last.transform(one)["verdict"].iloc[0])
finally:
    srv.stop()
