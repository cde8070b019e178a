#!/bin/bash
# Training driver — role parity with the reference's train.sh (background
# single-device run). On MI355X, multi-GPU data parallel is launched with
# torchrun, one rank per GPU over RCCL:
#   NGPU=8 bash scripts/train.sh
set -e
cd "$(dirname "$0")/.."

NGPU=${NGPU:-1}
export HSA_ENABLE_IPC_MODE_LEGACY=${HSA_ENABLE_IPC_MODE_LEGACY:-0}

if [ "$NGPU" -gt 1 ]; then
  python -m torch.distributed.run --nnodes=1 --nproc-per-node "$NGPU" \
    --master-addr 127.0.0.1 --master-port 29517 \
    scripts/train_nats.py > log.txt 2>&1 &
else
  python -u scripts/train_nats.py > log.txt 2>&1 &
fi
echo "training started (NGPU=$NGPU); tail -f log.txt"
