#!/bin/bash
# mixed_sync scenario (GeoMX scripts/*/run_mixed_sync.sh analog)
source "$(dirname "$0")/common.sh"
run_example cnn.py --mixed-sync "$@"
