#!/bin/bash
# Training launcher (parity: ESR:scripts/train_ours.sh — torch.distributed
# launch with per-node GPU count).  Usage:
#   bash scripts/train_esr.sh <nproc> <config.yml> [extra train.py args...]
set -e
NPROC=${1:-1}
CONFIG=${2:-configs/train_synth_2x.yml}
shift 2 || true
if [ "$NPROC" -gt 1 ]; then
  exec python -m torch.distributed.run --nnodes=1 --nproc-per-node "$NPROC" \
      --master-addr 127.0.0.1 train.py -c "$CONFIG" "$@"
fi
exec python train.py -c "$CONFIG" "$@"
