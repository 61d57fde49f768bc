#!/usr/bin/env python3
"""HiPS-vs-flat speedup experiment under an emulated WAN cap.

Runs bench.py across a config matrix via torchrun (one rank per
GPU/CPU-proc) and reports the speedup of HiPS+compression over the flat
all_reduce baseline at identical emulated inter-DC bandwidth — the
experiment behind the reference's "20x acceleration under identical
network bandwidth conditions" claim (README.md:12).

  python scripts/hips_vs_flat.py --nproc 8 --wan-gbps 1 --steps 10

Writes a markdown table to stdout and JSON rows to --json-out.
"""

import argparse
import json
import os
import subprocess
import sys

HERE = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run_bench(nproc, extra, steps, warmup, bs, port):
    out = os.path.join(HERE, "gpurun_out", f"hvf_{port}.json")
    os.makedirs(os.path.dirname(out), exist_ok=True)
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           f"--nproc-per-node={nproc}", "--master-addr", "127.0.0.1",
           "--master-port", str(port), os.path.join(HERE, "bench.py"),
           "--gpus", str(nproc), "--steps", str(steps),
           "--warmup", str(warmup), "--batch-size", str(bs),
           "--json-out", out] + extra
    r = subprocess.run(cmd, capture_output=True, text=True, timeout=1800)
    if r.returncode != 0:
        print(r.stdout[-4000:], file=sys.stderr)
        print(r.stderr[-4000:], file=sys.stderr)
        raise RuntimeError(f"bench failed: {extra}")
    with open(out) as f:
        return json.loads(f.read())


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--nproc", type=int, default=4)
    ap.add_argument("--parties", type=int, default=2)
    ap.add_argument("--wan-gbps", type=float, default=1.0)
    ap.add_argument("--steps", type=int, default=8)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--batch-size", type=int, default=16)
    ap.add_argument("--image-size", type=int, default=28)
    ap.add_argument("--bsc-ratio", type=float, default=0.01)
    ap.add_argument("--port", type=int, default=29640)
    ap.add_argument("--backend", type=str, default="auto",
                    help="pass 'gloo' to run N ranks on one GPU "
                         "(GPU compute, gloo wire; RCCL rejects "
                         "duplicate devices)")
    ap.add_argument("--json-out", type=str, default=None)
    args = ap.parse_args()

    common = ["--image-size", str(args.image_size),
              "--parties", str(args.parties),
              "--wan-gbps", str(args.wan_gbps),
              "--backend", args.backend]
    configs = [
        ("flat (baseline)", ["--mode", "flat"]),
        ("hips dense", ["--mode", "hips"]),
        ("hips fp16", ["--mode", "hips", "--compress", "fp16"]),
        ("hips bsc", ["--mode", "hips", "--compress", "bsc",
                      "--bsc-ratio", str(args.bsc_ratio)]),
        ("hips mpq", ["--mode", "hips", "--compress", "mpq",
                      "--bsc-ratio", str(args.bsc_ratio)]),
        ("hips 2bit", ["--mode", "hips", "--compress", "2bit"]),
        ("hips dgt", ["--mode", "hips", "--compress", "dgt"]),
        ("hips async (stale-1)", ["--mode", "hips", "--sync-mode",
                                  "dist_async"]),
        ("hips async+fp16", ["--mode", "hips", "--sync-mode", "dist_async",
                             "--compress", "fp16"]),
    ]
    rows = []
    port = args.port
    for name, extra in configs:
        res = run_bench(args.nproc, common + extra, args.steps, args.warmup,
                        args.batch_size, port)
        port += 1
        rows.append((name, res))
        print(f"  {name:16s} {res['value']:12.1f} samples/s "
              f"{res['ms_per_step']:9.2f} ms/step", flush=True)

    base = rows[0][1]["value"]
    print(f"\nWAN cap {args.wan_gbps} Gbit/s, {args.nproc} ranks, "
          f"{args.parties} parties, model geomx_cnn "
          f"{args.image_size}px bs{args.batch_size}:\n")
    print(f"| config | samples/s | ms/step | speedup vs flat |")
    print(f"|---|---|---|---|")
    for name, res in rows:
        print(f"| {name} | {res['value']:.1f} | {res['ms_per_step']:.2f} "
              f"| {res['value']/base:.2f}x |")
    if args.json_out:
        with open(args.json_out, "w") as f:
            json.dump({"args": vars(args),
                       "rows": [{"name": n, **r} for n, r in rows]}, f,
                      indent=1)


if __name__ == "__main__":
    main()
