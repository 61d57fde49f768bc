set -x
cd /root/repo
mkdir -p gpurun_out
echo "== bench: split+find1+hipgraph (new default) ==" > gpurun_out/c6.log
timeout 700 python bench.py --steps 12 --warmup 4 2>gpurun_out/c6a.err | tee -a gpurun_out/c6.log
echo "== bench: no hip-graph (A/B) ==" >> gpurun_out/c6.log
timeout 500 python bench.py --steps 12 --warmup 4 --no-hip-graph 2>/dev/null | tee -a gpurun_out/c6.log
echo "== resnet50 1-GPU (config 5 base row) ==" >> gpurun_out/c6.log
timeout 700 python bench.py --model resnet50 --batch-size 256 --steps 8 --warmup 3 2>gpurun_out/c6b.err | tee -a gpurun_out/c6.log
echo DONE_C6 | tee -a gpurun_out/c6.log
