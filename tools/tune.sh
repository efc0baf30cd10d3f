#!/usr/bin/env bash
# LR grid harness (reference parity: src/tune.sh — 7 LR candidates x
# max_tuning_step steps, then regex-average the worker losses per trial via
# the tuning parser). Single MI355X node, torchrun instead of mpirun.
#
#   tools/tune.sh [NPROC] [extra distributed_nn flags...]
set -euo pipefail
cd "$(dirname "$0")/.."

NPROC="${1:-2}"
shift || true
TUNE_DIR="${TUNE_DIR:-output/tune}"
MAX_TUNING_STEP="${MAX_TUNING_STEP:-100}"
mkdir -p "$TUNE_DIR"
export HSA_ENABLE_IPC_MODE_LEGACY="${HSA_ENABLE_IPC_MODE_LEGACY:-0}"

echo "Start parameter tuning ..."
for lr in 0.0078125 0.015625 0.03125 0.0625 0.125 0.25 0.5; do
  echo "Trial running for learning rate: ${lr}"
  log="$TUNE_DIR/lr_${lr}.log"
  python -m torch.distributed.run --nnodes=1 --nproc-per-node "$NPROC" \
      --master-addr 127.0.0.1 --master-port "${MASTER_PORT:-29512}" \
      -m ps_pytorch_amd.distributed_nn \
      --lr="$lr" --momentum=0.9 \
      --network=ResNet18 --dataset=Cifar10 \
      --batch-size=8 --test-batch-size=200 \
      --comm-type=Bcast --num-aggregate="$((NPROC - 1))" \
      --max-steps="$MAX_TUNING_STEP" --eval-freq=1000000 \
      --compress-grad=compress --enable-gpu=true --log-interval=10 \
      "$@" > "$log" 2>&1 || true
  python -m ps_pytorch_amd.tuning_parser --tuning-dir "$log" \
      --tuning-lr "$lr" --num-workers "$((NPROC - 1))" \
      --max-tuning-step "$MAX_TUNING_STEP"
done
