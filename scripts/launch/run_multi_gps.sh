#!/bin/bash
# multi_gps scenario (GeoMX scripts/*/run_multi_gps.sh analog)

# MultiGPS: sharded global tier across party leaders; use PARTIES>=2
PARTIES=${PARTIES:-4}
source "$(dirname "$0")/common.sh"
run_example cnn.py --global-mode sharded "$@"
