#!/usr/bin/env bash
# Launch synchronous PS training on ONE MI355X node (reference parity:
# src/run_pytorch.sh, which did `mpirun -n N --hostfile hosts_address`).
# MI355X-native launch is one rank per GPU over RCCL via torchrun; rank 0
# becomes the PS, ranks 1..N-1 the workers.
#
#   tools/run_pytorch.sh [NPROC] [extra distributed_nn flags...]
#
# Writes the torchrun PID to output/run_pytorch.pid (tools/killall.sh stops it).
set -euo pipefail
cd "$(dirname "$0")/.."

NPROC="${1:-$(python -c 'import torch;print(max(torch.cuda.device_count(),2))')}"
shift || true
mkdir -p output
export HSA_ENABLE_IPC_MODE_LEGACY="${HSA_ENABLE_IPC_MODE_LEGACY:-0}"

python -m torch.distributed.run --nnodes=1 --nproc-per-node "$NPROC" \
    --master-addr 127.0.0.1 --master-port "${MASTER_PORT:-29500}" \
    -m ps_pytorch_amd.distributed_nn \
    --lr=0.1 \
    --momentum=0.9 \
    --network=ResNet18 \
    --dataset=Cifar10 \
    --batch-size=128 \
    --test-batch-size=1000 \
    --comm-type=Bcast \
    --num-aggregate=5 \
    --mode=normal \
    --eval-freq=50 \
    --epochs=10 \
    --max-steps=1000000 \
    --compress-grad=compress \
    --enable-gpu=true \
    --train-dir=output/models/ \
    "$@" &
echo $! > output/run_pytorch.pid
wait
