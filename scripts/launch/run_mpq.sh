#!/bin/bash
# mpq scenario (GeoMX scripts/*/run_mpq.sh analog)
source "$(dirname "$0")/common.sh"
run_example cnn_mpq.py "$@"
