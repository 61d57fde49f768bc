#!/usr/bin/env python3
"""Bucket-size x comm-dtype sweep for the flat DP path (ROADMAP item 4).

xGMI links are point-to-point (7 x ~153 GB/s per GPU): several
in-flight ring all_reduces stripe across links, one giant bucket
serializes on a single ring. This sweep finds the knee on a real
multi-GPU node.

  python scripts/bucket_sweep.py --nproc 8 --steps 15
"""

import argparse
import json
import os
import subprocess
import sys

HERE = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run(nproc, bucket_mb, comm_dtype, steps, warmup, bs, port):
    out = os.path.join(HERE, "gpurun_out", f"bsw_{bucket_mb}_{comm_dtype}.json")
    os.makedirs(os.path.dirname(out), exist_ok=True)
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           f"--nproc-per-node={nproc}", "--master-addr", "127.0.0.1",
           "--master-port", str(port), os.path.join(HERE, "bench.py"),
           "--gpus", str(nproc), "--steps", str(steps),
           "--warmup", str(warmup), "--batch-size", str(bs),
           "--bucket-mb", str(bucket_mb), "--comm-dtype", comm_dtype,
           "--json-out", out]
    r = subprocess.run(cmd, capture_output=True, text=True, timeout=1800)
    if r.returncode != 0:
        print(r.stdout[-1500:], r.stderr[-1500:], file=sys.stderr)
        raise RuntimeError(f"bench failed: bucket={bucket_mb} {comm_dtype}")
    with open(out) as f:
        return json.loads(f.read())


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--nproc", type=int, default=8)
    ap.add_argument("--steps", type=int, default=15)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--batch-size", type=int, default=512)
    ap.add_argument("--buckets", type=str, default="5,10,25,50,100")
    ap.add_argument("--dtypes", type=str, default="fp32,bf16")
    ap.add_argument("--port", type=int, default=29750)
    ap.add_argument("--json-out", type=str, default=None)
    args = ap.parse_args()

    rows = []
    port = args.port
    print(f"\nbucket sweep, {args.nproc} ranks, bs{args.batch_size}\n")
    print("| bucket MB | comm dtype | samples/s | ms/step |")
    print("|---|---|---|---|")
    for mb in [int(x) for x in args.buckets.split(",")]:
        for cd in args.dtypes.split(","):
            res = run(args.nproc, mb, cd, args.steps, args.warmup,
                      args.batch_size, port)
            port += 2
            print(f"| {mb} | {cd} | {res['value']:.0f} | "
                  f"{res['ms_per_step']:.2f} |", flush=True)
            rows.append({"bucket_mb": mb, "comm_dtype": cd, **res})
    if args.json_out:
        with open(args.json_out, "w") as f:
            json.dump(rows, f, indent=1)


if __name__ == "__main__":
    main()
