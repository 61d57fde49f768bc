#!/bin/bash
# fp16 scenario (GeoMX scripts/*/run_fp16.sh analog)
source "$(dirname "$0")/common.sh"
run_example cnn_fp16.py "$@"
