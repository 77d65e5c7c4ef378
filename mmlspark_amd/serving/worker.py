"""Standalone serving worker process — `python -m mmlspark_amd.serving.worker`.

The process analog of the reference's per-executor WorkerServer
(HTTPSourceV2.scala:475): loads a saved Transformer (or a saved GBDT model
for the low-latency scorer path), serves it over HTTP, and POSTs its
ServiceInfo to a head/driver discovery URL (reportServerToDriver,
HTTPSourceV2.scala:670-676).  One worker per GPU rank in a real
deployment (HIP_VISIBLE_DEVICES pins the device).
"""
from __future__ import annotations

import argparse
import json
import signal
import sys
import time


def main(argv=None):
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", required=True,
                    help="saved stage directory (core.serialize format)")
    ap.add_argument("--output-cols", default="prediction",
                    help="comma-separated columns returned per request")
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--port", type=int, default=0)
    ap.add_argument("--mode", default="micro-batch",
                    choices=["micro-batch", "continuous"])
    ap.add_argument("--name", default="mmlspark-worker")
    ap.add_argument("--report-to", default=None,
                    help="head discovery URL to POST ServiceInfo to")
    ap.add_argument("--scorer", action="store_true",
                    help="serve through LowLatencyGBDTScorer (GBDT models)")
    args = ap.parse_args(argv)

    import mmlspark_amd
    mmlspark_amd._register_all()  # load_stage needs the full registry
    from ..core.serialize import load_stage
    from .server import LowLatencyGBDTScorer, ServingServer, TransformerHandler

    stage = load_stage(args.model)
    if args.scorer:
        handler = LowLatencyGBDTScorer(stage.booster)
    else:
        handler = TransformerHandler(stage, args.output_cols.split(","))
    srv = ServingServer(handler, host=args.host, port=args.port,
                        mode=args.mode, name=args.name).start()
    # stdout line lets a parent process learn the bound port
    print(json.dumps({"ready": True, **srv.service_info()}), flush=True)
    if args.report_to:
        import requests
        try:
            requests.post(args.report_to,
                          json={"__register__": srv.service_info()},
                          timeout=10)
        except Exception as e:  # discovery is best-effort
            print(json.dumps({"report_error": repr(e)}), flush=True)

    stop = {"flag": False}

    def _sig(*_):
        stop["flag"] = True

    signal.signal(signal.SIGTERM, _sig)
    signal.signal(signal.SIGINT, _sig)
    while not stop["flag"]:
        time.sleep(0.2)
    srv.stop()
    return 0


if __name__ == "__main__":
    sys.exit(main())
