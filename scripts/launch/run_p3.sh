#!/bin/bash
# p3 scenario (GeoMX scripts/*/run_p3.sh analog)
# P3 key-slicing engages automatically for keys >= MXNET_KVSTORE_BIGARRAY_BOUND
export MXNET_KVSTORE_BIGARRAY_BOUND=${MXNET_KVSTORE_BIGARRAY_BOUND:-100000}

source "$(dirname "$0")/common.sh"
run_example cnn.py "$@"
