#!/bin/bash
# Shared launcher: pseudo-distributed HiPS topology on one host
# (the scripts/{cpu,gpu}/run_*.sh analog — one torchrun replaces the
# reference's 12-process scheduler/server/worker spawn).
NPROC=${NPROC:-4}
PARTIES=${PARTIES:-2}
PORT=${PORT:-29500}
EXTRA=${EXTRA:-}
run_example() {
  local script=$1; shift
  python -m torch.distributed.run --nnodes=1 --nproc-per-node "$NPROC" \
    --master-addr 127.0.0.1 --master-port "$PORT" \
    "$(dirname "$0")/../../examples/$script" --parties "$PARTIES" "$@" $EXTRA
}
