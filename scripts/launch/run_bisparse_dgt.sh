#!/bin/bash
# BASELINE config 5 composition: Bi-Sparse content selection + DGT
# 4-bit value tier on the kvstore WAN exchange.
# Usage: bash scripts/launch/run_bisparse_dgt.sh [NPROC] [PARTIES]
set -euo pipefail
cd "$(dirname "$0")/../.."
NPROC=${1:-4}
PARTIES=${2:-2}
PORT=${MASTER_PORT:-$((29500 + RANDOM % 400))}
python -m torch.distributed.run --nnodes=1 --nproc-per-node "$NPROC" \
  --master-addr 127.0.0.1 --master-port "$PORT" \
  examples/cnn_bsc_dgt.py --parties "$PARTIES" -lr 0.001 \
  --max-iters "${MAX_ITERS:-20}" "$@"
