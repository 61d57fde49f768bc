set -x
cd /root/repo
mkdir -p gpurun_out
echo "== convergence (kvstore+custom kernels, GPU, synthetic) ==" > gpurun_out/c12.log
timeout 500 python examples/cnn.py --image-size 224 --batch-size 64 \
  --data-n 2048 --max-iters 40 --measure-out gpurun_out/r02_cnn_convergence.jsonl \
  2>/dev/null | tail -6 | tee -a gpurun_out/c12.log
echo "== bsc convergence (update-on-worker) ==" >> gpurun_out/c12.log
timeout 400 python examples/cnn_bsc.py --image-size 224 --batch-size 64 \
  --data-n 2048 --max-iters 25 2>/dev/null | tail -3 | tee -a gpurun_out/c12.log
echo "== TSEngine on HW (2 ranks, heterogeneous uplinks) ==" >> gpurun_out/c12.log
port=$((29950 + RANDOM % 40))
ENABLE_INTER_TS=1 timeout 420 python -m torch.distributed.run --nnodes=1 \
  --nproc-per-node 2 --master-addr 127.0.0.1 --master-port $port \
  bench.py --gpus 2 --steps 4 --warmup 1 --batch-size 128 --backend gloo \
  --mode hips --parties 2 --party-wan-gbps 1,0.25 2>/dev/null | tail -1 | tee -a gpurun_out/c12.log
port=$((29990 + RANDOM % 9))
timeout 420 python -m torch.distributed.run --nnodes=1 \
  --nproc-per-node 2 --master-addr 127.0.0.1 --master-port $port \
  bench.py --gpus 2 --steps 4 --warmup 1 --batch-size 128 --backend gloo \
  --mode hips --parties 2 --party-wan-gbps 1,0.25 2>/dev/null | tail -1 | tee -a gpurun_out/c12.log
echo DONE_C12 | tee -a gpurun_out/c12.log
