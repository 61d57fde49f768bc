set -x
cd /root/repo
mkdir -p gpurun_out
echo "== gpu suite (incl tr16/wrw2/bsc_fused) ==" > gpurun_out/c4.log
timeout 900 python -m pytest tests -m gpu -q 2>&1 | tail -25 | tee -a gpurun_out/c4.log
echo "== kernel bench ==" >> gpurun_out/c4.log
timeout 300 python scripts/kernel_bench.py 2>&1 | tee gpurun_out/c4_kernel_bench.txt | tail -20 >> gpurun_out/c4.log
echo "== bench default ==" >> gpurun_out/c4.log
timeout 600 python bench.py --steps 10 --warmup 3 2>/dev/null | tee -a gpurun_out/c4.log
echo "== bench split-bwd (custom wrw v2 on conv2) ==" >> gpurun_out/c4.log
GEOPS_SPLIT_BWD=1 timeout 600 python bench.py --steps 10 --warmup 3 2>/dev/null | tee -a gpurun_out/c4.log
echo DONE_C4 | tee -a gpurun_out/c4.log
