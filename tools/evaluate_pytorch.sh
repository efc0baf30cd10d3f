#!/usr/bin/env bash
# Launch the polling evaluator against the training run's checkpoint dir
# (reference parity: src/evaluate_pytorch.sh + distributed_evaluator.py —
# a separate non-distributed process that shares only the model directory).
set -euo pipefail
cd "$(dirname "$0")/.."

python -m ps_pytorch_amd.evaluator \
    --network=ResNet18 \
    --dataset=Cifar10 \
    --test-batch-size=1000 \
    --train-dir=output/models/ \
    --eval-freq=50 \
    "$@"
