# Speedup-vs-WAN-cap curve: flat vs hips{bsc,mpq,fp16} at 0.5 and 0.2 Gbit/s
set -x
cd /root/repo
mkdir -p gpurun_out
for gb in 0.5 0.2; do
python - <<PY 2>&1 | tee -a gpurun_out/hvf_curve.log
import json, subprocess, sys, os
gb = $gb if False else float("$gb")
rows = []
port = 29700
for name, extra in [
    ("flat", ["--mode", "flat"]),
    ("hips bsc", ["--mode", "hips", "--compress", "bsc"]),
    ("hips mpq", ["--mode", "hips", "--compress", "mpq"]),
    ("hips fp16", ["--mode", "hips", "--compress", "fp16"]),
]:
    out = f"gpurun_out/hvfc_{gb}_{name.replace(' ','_')}.json"
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           "--nproc-per-node=4", "--master-addr", "127.0.0.1",
           "--master-port", str(port), "bench.py", "--gpus", "4",
           "--steps", "4", "--warmup", "1", "--batch-size", "128",
           "--image-size", "224", "--backend", "gloo", "--parties", "2",
           "--wan-gbps", str(gb), "--json-out", out] + extra
    port += 1
    r = subprocess.run(cmd, capture_output=True, text=True, timeout=1500)
    if r.returncode != 0:
        print("FAIL", name, r.stderr[-1500:]); continue
    d = json.load(open(out))
    rows.append((name, d))
    print(f"wan={gb} {name:10s} {d['value']:10.1f} samples/s {d['ms_per_step']:9.2f} ms/step", flush=True)
if rows:
    base = rows[0][1]["value"]
    for n, d in rows:
        print(f"| {gb} | {n} | {d['value']:.1f} | {d['ms_per_step']:.2f} | {d['value']/base:.2f}x |")
PY
done
echo DONE_C3
