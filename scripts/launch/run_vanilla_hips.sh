#!/bin/bash
# vanilla_hips scenario (GeoMX scripts/*/run_vanilla_hips.sh analog)
source "$(dirname "$0")/common.sh"
run_example cnn.py "$@"
