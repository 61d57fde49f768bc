set -x
cd /root/repo
mkdir -p gpurun_out
echo "== convergence (GPU, find-mode 3, small shape) ==" > gpurun_out/c14.log
MIOPEN_FIND_MODE=3 timeout 420 python examples/cnn.py --image-size 64 \
  --batch-size 64 --data-n 2048 --max-iters 30 \
  --measure-out gpurun_out/r02_cnn_convergence.jsonl \
  > gpurun_out/c14_conv.log 2>&1
tail -4 gpurun_out/c14_conv.log | tee -a gpurun_out/c14.log
MIOPEN_FIND_MODE=3 timeout 300 python examples/cnn_bsc.py --image-size 64 \
  --batch-size 64 --data-n 2048 --max-iters 15 > gpurun_out/c14_bsc.log 2>&1
tail -2 gpurun_out/c14_bsc.log | tee -a gpurun_out/c14.log
echo "== final profile ==" >> gpurun_out/c14.log
cd /tmp && export TMPDIR=/tmp && cd /root/repo
timeout 500 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof14 -o c14 -- python bench.py --steps 8 --warmup 3 > gpurun_out/c14_prof.log 2>&1
echo PROF_RC=$? | tee -a gpurun_out/c14.log
echo DONE_C14 | tee -a gpurun_out/c14.log
