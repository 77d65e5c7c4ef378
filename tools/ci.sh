#!/usr/bin/env bash
# CI matrix — the per-package UnitTests analog of the reference's
# pipeline.yaml:321-404: one job per package split, a flaky retry loop
# (pipeline.yaml:390-404), and per-job timeouts.
#
# Usage:
#   tools/ci.sh                # run every split
#   tools/ci.sh gbdt serving   # run selected splits
#   CI_GPU=1 tools/ci.sh gpu   # GPU split (on an MI355X box)
set -u
cd "$(dirname "$0")/.."

RETRIES=${CI_RETRIES:-2}          # flaky retry loop
TIMEOUT=${CI_TIMEOUT:-1200}       # seconds per split (ref: 20-min sbt cap)

declare -A SPLITS=(
  [core]="tests/test_core.py tests/test_io_codegen.py tests/test_interop.py tests/test_r_bindings.py"
  [stages]="tests/test_stages.py tests/test_properties.py tests/test_fuzzing.py tests/test_text.py tests/test_automl.py tests/test_coverage_extras.py"
  [gbdt]="tests/test_gbdt.py tests/test_gbdt_sparse.py tests/test_benchmarks_csv.py tests/test_external_anchor.py tests/test_elastic.py"
  [vw]="tests/test_vw.py"
  [distributed]="tests/test_distributed.py"
  [serving]="tests/test_serving.py tests/test_serving_cluster_remote.py"
  [cognitive]="tests/test_images_cognitive.py tests/test_jpeg_native.py"
  [models]="tests/test_models_misc.py tests/test_cyber.py tests/test_explainers.py tests/test_deep_learning.py tests/test_knn.py tests/test_sar.py tests/test_image_featurizer.py"
  [e2e]="tests/test_integration_e2e.py tests/test_examples_e2e.py"
  [gpu]="tests -m gpu"
)
ORDER=(core stages gbdt vw distributed serving cognitive models e2e)

run_split() {
  local name="$1"; shift
  local args=(${SPLITS[$name]})
  local attempt=0
  while :; do
    attempt=$((attempt + 1))
    echo "=== [$name] attempt $attempt ==="
    if timeout "$TIMEOUT" python -m pytest -x -q "${args[@]}"; then
      echo "=== [$name] PASSED ==="
      return 0
    fi
    if [ "$attempt" -gt "$RETRIES" ]; then
      echo "=== [$name] FAILED after $attempt attempts ==="
      return 1
    fi
    echo "=== [$name] retrying (flaky loop) ==="
  done
}

targets=("$@")
if [ ${#targets[@]} -eq 0 ]; then
  targets=("${ORDER[@]}")
  [ "${CI_GPU:-0}" = "1" ] && targets+=(gpu)
fi

fail=0
for t in "${targets[@]}"; do
  run_split "$t" || fail=1
done
exit $fail
