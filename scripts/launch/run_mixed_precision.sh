#!/bin/bash
# mixed-precision scenario (GeoMX scripts/*/run_mixed_precision.sh —
# same cnn_mpq.py driver: MPQ size-gates fp16/bsc wires over an fp32
# master; compute precision is the trainer's autocast dtype)
source "$(dirname "$0")/common.sh"
run_example cnn_mpq.py "$@"
