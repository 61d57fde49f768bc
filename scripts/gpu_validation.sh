#!/bin/bash
# One-call GPU validation sweep — run on a GPU box (e.g. under gpurun):
#   gpurun --timeout 1500 -- 'bash scripts/gpu_validation.sh'
# Covers: the gpu test suite, the flagship bench, a kernel bandwidth
# table, and 2-process distributed smokes (gloo keeps tensors on CPU —
# these validate the DISTRIBUTED LOGIC of flat, HiPS+BSC under a WAN
# cap, and the TSEngine relay on the box; GPU compute itself is
# covered by the gpu suite and the 1-proc bench above) before the
# driver's multi-GPU RCCL scaling run.
set -euo pipefail
cd "$(dirname "$0")/.."
mkdir -p gpurun_out

echo "== gpu test suite =="
python -m pytest tests -m gpu -x -q 2>&1 | tail -3

echo "== flagship bench (1 GPU) =="
python bench.py --steps 10 --warmup 3 | tee gpurun_out/val_bench.json

echo "== kernel bandwidths =="
python scripts/kernel_bench.py | tee gpurun_out/val_kernel_bench.txt

echo "== 2-proc distributed smokes (gloo, 1 GPU shared) =="
for args in \
  "--mode flat" \
  "--mode hips --parties 2 --compress bsc --wan-gbps 0.5" \
  "--mode hips --parties 2 --wan-gbps 0.5"; do
  port=$((29800 + RANDOM % 100))
  timeout 600 python -m torch.distributed.run --nnodes=1 \
    --nproc-per-node 2 --master-addr 127.0.0.1 --master-port $port \
    bench.py --gpus 2 --steps 3 --warmup 1 --batch-size 32 \
    --backend gloo $args | tail -1
done
port=$((29900 + RANDOM % 100))
ENABLE_INTER_TS=1 timeout 600 python -m torch.distributed.run --nnodes=1 \
  --nproc-per-node 2 --master-addr 127.0.0.1 --master-port $port \
  bench.py --gpus 2 --steps 3 --warmup 1 --batch-size 32 \
  --backend gloo --mode hips --parties 2 \
  --party-wan-gbps 1,0.2 | tail -1

echo "== smoke() =="
python -c "import __graft_entry__ as g; g.smoke(); print('smoke ok')"
echo "ALL GPU VALIDATION PASSED"
