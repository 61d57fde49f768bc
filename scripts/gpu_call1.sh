set -x
cd /root/repo
mkdir -p gpurun_out
echo "== gpu suite ==" | tee gpurun_out/c1.log
python -m pytest tests -m gpu -x -q 2>&1 | tail -4 | tee -a gpurun_out/c1.log
echo "== bench ==" | tee -a gpurun_out/c1.log
python bench.py --steps 10 --warmup 3 2>gpurun_out/bench.err | tee -a gpurun_out/c1.log
echo "== rccl 2-rank 1-gpu probe ==" | tee -a gpurun_out/c1.log
timeout 240 python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
  --master-addr 127.0.0.1 --master-port 29611 scripts/rccl_probe.py \
  2>&1 | tail -5 | tee gpurun_out/probe.log
if grep -q RCCL_PROBE_OK gpurun_out/probe.log; then
  echo "== 2-rank hips bsc smoke over RCCL ==" | tee -a gpurun_out/c1.log
  timeout 600 python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
    --master-addr 127.0.0.1 --master-port 29612 bench.py --gpus 2 --steps 3 \
    --warmup 1 --batch-size 64 --mode hips --parties 2 --compress bsc \
    --wan-gbps 0.5 2>&1 | tail -3 | tee -a gpurun_out/c1.log
fi
echo DONE_C1 | tee -a gpurun_out/c1.log
