#!/usr/bin/env bash
# Integration runner for the pytorch-flavor examples
# (the reference's examples/run_pytorch_examples.sh analog).
set -e
cd "$(dirname "$0")/pytorch"

for ex in pytorch_example.py pytorch_distributed_example.py; do
    echo "=== running $ex ==="
    MODEL_DIR="$(mktemp -d)" timeout 300 python "$ex"
    echo "=== $ex OK ==="
done
echo "all pytorch examples passed"
