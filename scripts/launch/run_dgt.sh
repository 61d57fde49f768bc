#!/bin/bash
# dgt scenario (GeoMX scripts/*/run_dgt.sh analog)
source "$(dirname "$0")/common.sh"
run_example cnn_dgt.py "$@"
