#!/usr/bin/env bash
# One-shot reproduction of every headline number -> JSON lines on stdout.
# Usage (GPU box): bash tools/run_all_benches.sh > gpurun_out/all_benches.jsonl
set -u
cd "$(dirname "$0")/.."
run() { timeout "${2:-420}" python $1 2>/dev/null | grep "^{" | tail -1; }
run "bench.py --steps 30 --warmup 5"
run "bench.py --steps 15 --warmup 3 --categorical 4"
run "bench.py --steps 10 --warmup 3 --sparse"
run "bench.py --rows 100000000 --steps 10 --warmup 2" 600
run "bench_vw.py"
run "bench_shap.py --rows 64"
run "bench_serving.py --requests 2000"
run "bench_serving.py --requests 3200 --clients 64" 600
run "bench_image.py --steps 20 --warmup 5"
run "bench_image.py --bytes-to-features" 600
timeout 600 python tools/bench_misc.py 2>/dev/null | grep "^{"
