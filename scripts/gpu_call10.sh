set -x
cd /root/repo
mkdir -p gpurun_out
echo "== full gpu suite ==" > gpurun_out/c10.log
timeout 600 python -m pytest tests -m gpu -q 2>&1 | tail -4 | tee -a gpurun_out/c10.log
echo "== bench ==" >> gpurun_out/c10.log
timeout 420 python bench.py --steps 12 --warmup 4 2>/dev/null | tail -1 | tee -a gpurun_out/c10.log
echo "== smoke ==" >> gpurun_out/c10.log
timeout 300 python -c "import __graft_entry__ as g; g.smoke()" 2>&1 | tail -1 | tee -a gpurun_out/c10.log
echo "== sanitize (quick) ==" >> gpurun_out/c10.log
timeout 420 bash scripts/sanitize.sh 1 2>&1 | tail -8 | tee -a gpurun_out/c10.log
echo DONE_C10 | tee -a gpurun_out/c10.log
