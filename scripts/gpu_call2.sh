# HiPS-vs-flat matrix on MI355X: GPU compute, gloo wire, real payloads,
# full-size flagship config under a 1 Gbit/s emulated WAN cap.
set -x
cd /root/repo
mkdir -p gpurun_out
python scripts/hips_vs_flat.py --backend gloo --nproc 4 --parties 2 --wan-gbps 1.0 \
  --steps 8 --warmup 2 --batch-size 128 --image-size 224 \
  --json-out gpurun_out/hips_vs_flat_gpu_ws4.json 2>&1 | tee gpurun_out/hvf4.log
echo DONE_C2
