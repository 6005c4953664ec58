#!/usr/bin/env bash
# Integration runner (the reference's examples/run_examples.sh analog):
# run every estimator/keras example and assert exit codes.
set -e
cd "$(dirname "$0")"
export MIYARN_APP_BASE_DIR="${MIYARN_APP_BASE_DIR:-/tmp/miyarn_examples}"

for ex in keras_example.py allreduce_example.py dnn_classifier_example.py \
          linear_classifier_example.py keras_wide_deep_example.py \
          mlflow_example.py wide_deep_example.py; do
    echo "=== running $ex ==="
    MODEL_DIR="$(mktemp -d)" timeout 300 python "$ex"
    echo "=== $ex OK ==="
done
echo "all examples passed"
