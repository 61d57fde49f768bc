#!/bin/bash
# One-call GPU validation sweep — run on a GPU box (e.g. under gpurun):
#   gpurun --timeout 1800 -- 'bash scripts/gpu_validation.sh'
# Covers: the full GPU suite, smoke, the flagship bench, kernel
# bandwidths, the sanitizer lane, and 2-process distributed smokes
# (gloo wire + CUDA compute on a single GPU).
set -euo pipefail
cd "$(dirname "$0")/.."
mkdir -p gpurun_out

echo "== gpu test suite =="
python -m pytest tests -m gpu -x -q 2>&1 | tail -3

echo "== smoke =="
python -c "import __graft_entry__ as g; g.smoke(); print('smoke ok')"

echo "== flagship bench (1 GPU) =="
python bench.py --steps 12 --warmup 4 | tee gpurun_out/val_bench.json

echo "== kernel bandwidths =="
python scripts/kernel_bench.py | tee gpurun_out/val_kernel_bench.txt

echo "== sanitizer lane =="
bash scripts/sanitize.sh 2 | tail -4

echo "== 2-proc distributed smokes (gloo wire, CUDA compute) =="
for args in \
  "--mode flat" \
  "--mode hips --parties 2 --compress bsc --wan-gbps 0.5" \
  "--mode hips --parties 2 --compress bsc_dgt --wan-gbps 0.5"; do
  port=$((29800 + RANDOM % 100))
  timeout 600 python -m torch.distributed.run --nnodes=1 \
    --nproc-per-node 2 --master-addr 127.0.0.1 --master-port $port \
    bench.py --gpus 2 --steps 3 --warmup 1 --batch-size 64 \
    --backend gloo $args | tail -1
done
echo "ALL GPU VALIDATION PASSED"
