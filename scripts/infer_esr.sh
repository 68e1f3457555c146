#!/bin/bash
# Inference launcher (parity: ESR:scripts/infer_ours.sh — checkpoint +
# dataset flags -> ESR-vs-bicubic tables).  Usage:
#   bash scripts/infer_esr.sh <checkpoint.pth> <data_path|datalist.txt> [out]
set -e
CKPT=${1:?checkpoint path required}
DATA=${2:?data path required}
OUT=${3:-runs/eval}
shift 3 || true
exec python infer.py --model_path "$CKPT" --data_path "$DATA" \
    --output_path "$OUT" --scale 2 --ori_scale down2 \
    --window 2048 --sliding_window 1024 --seql 5 --seqn 3 \
    --need_gt_events "$@"
