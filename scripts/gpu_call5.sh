set -x
cd /root/repo
mkdir -p gpurun_out
echo "== full gpu suite ==" > gpurun_out/c5.log
timeout 900 python -m pytest tests -m gpu -q 2>&1 | tail -6 | tee -a gpurun_out/c5.log
echo "== kernel bench (fused bsc rows) ==" >> gpurun_out/c5.log
timeout 420 python scripts/kernel_bench.py 2>&1 | tee gpurun_out/c5_kernel_bench.txt | tail -22 >> gpurun_out/c5.log
echo "== bench default ==" >> gpurun_out/c5.log
timeout 600 python bench.py --steps 12 --warmup 4 2>/dev/null | tee -a gpurun_out/c5.log
echo "== bench split-bwd ==" >> gpurun_out/c5.log
GEOPS_SPLIT_BWD=1 timeout 600 python bench.py --steps 12 --warmup 4 2>/dev/null | tee -a gpurun_out/c5.log
echo "== rocprof split-bwd steady ==" >> gpurun_out/c5.log
cd /tmp && export TMPDIR=/tmp && cd /root/repo
GEOPS_SPLIT_BWD=1 timeout 600 rocprofv3 --kernel-trace --stats -d gpurun_out/prof5 -o c5 -- python bench.py --steps 12 --warmup 4 > gpurun_out/c5_prof.log 2>&1
grep -A 30 "KERNEL_DISPATCH" $(ls gpurun_out/prof5/*stats* 2>/dev/null | head -1) 2>/dev/null | head -35 >> gpurun_out/c5.log || tail -20 gpurun_out/c5_prof.log >> gpurun_out/c5.log
echo DONE_C5 | tee -a gpurun_out/c5.log
