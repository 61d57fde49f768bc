#!/bin/bash
# bisparse scenario (GeoMX scripts/*/run_bisparse.sh analog)
source "$(dirname "$0")/common.sh"
run_example cnn_bsc.py "$@"
