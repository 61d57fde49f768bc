#!/bin/bash
# tsengine scenario (GeoMX scripts/*/run_tsengine.sh analog):
# ENABLE_INTER_TS=1 turns on the throughput-matrix relay scheduler for
# the leader tier (kvstore/tsengine.py). Pair with GEOMX_PARTY_WAN_GBPS
# (e.g. "1,1,1,0.1") to emulate the heterogeneous WAN it targets.


source "$(dirname "$0")/common.sh"
export ENABLE_INTER_TS=${ENABLE_INTER_TS:-1}
run_example cnn.py --global-mode replicated "$@"
