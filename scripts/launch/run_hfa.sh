#!/bin/bash
# hfa scenario (GeoMX scripts/*/run_hfa.sh analog)
source "$(dirname "$0")/common.sh"
run_example cnn_hfa.py "$@"
