#!/bin/bash
# dcasgd scenario (GeoMX scripts/*/run_dcasgd.sh analog)
source "$(dirname "$0")/common.sh"
run_example cnn.py --dcasgd "$@"
