#!/bin/bash
# Sanitizer / race-detection lane (the reference ships no
# sanitizers; this build adds what ROCm 7.2 offers without CUDA's
# compute-sanitizer):
#   1. serialized-launch pass: AMD_SERIALIZE_KERNEL=3 + HIP_LAUNCH_BLOCKING
#      surfaces async launch failures, OOB aborts and missing-wait races
#      at the faulting kernel instead of a later sync point;
#   2. race screen: the GPU numerics suite repeated R times — kernels with
#      missing s_waitcnt/barrier discipline fail intermittently, not
#      deterministically (multi-run screening per the CDNA4 guide);
#   3. leak check: HSA async copy serialization via HSA_ENABLE_SDMA=0
#      catches copies racing kernel writes.
# Run on a GPU box:  bash scripts/sanitize.sh [repeats]
set -euo pipefail
cd "$(dirname "$0")/.."
R=${1:-3}

echo "== pass 1: serialized launches (AMD_SERIALIZE_KERNEL=3) =="
AMD_SERIALIZE_KERNEL=3 HIP_LAUNCH_BLOCKING=1 \
  python -m pytest tests/test_kernels_gpu.py -q -x 2>&1 | tail -2

echo "== pass 2: race screen x${R} =="
for i in $(seq 1 "$R"); do
  python -m pytest tests/test_kernels_gpu.py tests/test_gpu_e2e.py -q -x \
    2>&1 | tail -1
done

echo "== pass 3: SDMA-off copy ordering =="
HSA_ENABLE_SDMA=0 python -m pytest tests/test_kernels_gpu.py -q -x \
  2>&1 | tail -2
echo "SANITIZE PASSED"
