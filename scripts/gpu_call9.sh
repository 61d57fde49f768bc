set -x
cd /root/repo
mkdir -p gpurun_out
echo "== conv/wrw tests ==" > gpurun_out/c9.log
timeout 300 python -m pytest tests/test_kernels_gpu.py -q -k "wrw or conv" 2>&1 | tail -3 | tee -a gpurun_out/c9.log
echo "== bench (wrw4 pipelined) ==" >> gpurun_out/c9.log
timeout 420 python bench.py --steps 12 --warmup 4 2>/dev/null | tail -1 | tee -a gpurun_out/c9.log
echo "== config 4: 4 parties x 1, fp16 + mpq @0.5Gbps ==" >> gpurun_out/c9.log
for cfg in "--compress fp16" "--compress mpq"; do
  port=$((29810 + RANDOM % 50))
  timeout 420 python -m torch.distributed.run --nnodes=1 --nproc-per-node 4 \
    --master-addr 127.0.0.1 --master-port $port bench.py --gpus 4 --steps 4 \
    --warmup 1 --batch-size 128 --backend gloo --mode hips --parties 4 \
    --wan-gbps 0.5 $cfg 2>/dev/null | tail -1 | tee -a gpurun_out/c9.log
done
echo "== config 5: resnet50 hips 2x1 + bsc_dgt @0.5Gbps ==" >> gpurun_out/c9.log
port=$((29870 + RANDOM % 50))
timeout 500 python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
  --master-addr 127.0.0.1 --master-port $port bench.py --gpus 2 --steps 4 \
  --warmup 1 --batch-size 64 --model resnet50 --backend gloo --mode hips \
  --parties 2 --wan-gbps 0.5 --compress bsc_dgt 2>/dev/null | tail -1 | tee -a gpurun_out/c9.log
port=$((29930 + RANDOM % 50))
timeout 500 python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
  --master-addr 127.0.0.1 --master-port $port bench.py --gpus 2 --steps 4 \
  --warmup 1 --batch-size 64 --model resnet50 --backend gloo --mode flat \
  --parties 2 --wan-gbps 0.5 2>/dev/null | tail -1 | tee -a gpurun_out/c9.log
echo DONE_C9 | tee -a gpurun_out/c9.log
