#!/bin/bash
# tsengine scenario (GeoMX scripts/*/run_tsengine.sh analog)


source "$(dirname "$0")/common.sh"
run_example cnn.py --global-mode replicated "$@"
