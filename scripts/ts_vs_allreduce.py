#!/usr/bin/env python3
"""TSEngine relay vs collective exchange under a HETEROGENEOUS WAN.

The reference's TSEngine exists for geo-topologies where one data
center's uplink is much slower than the others: a collective drags its
ring share across the slow link every round, while a scheduled relay
visits the slow node exactly once and learns to route around it.
This script measures that on the emulated WAN: same model, same
heterogeneous per-party rates (GEOMX_PARTY_WAN_GBPS), leader exchange
via plain collectives vs ENABLE_INTER_TS=1.

  python scripts/ts_vs_allreduce.py --nproc 4 --rates 1,1,1,0.1

Writes a markdown table to stdout and JSON rows to --json-out.
"""

import argparse
import json
import os
import subprocess
import sys

HERE = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run_bench(nproc, extra, steps, warmup, bs, port, env_extra):
    out = os.path.join(HERE, "gpurun_out", f"tsx_{port}.json")
    os.makedirs(os.path.dirname(out), exist_ok=True)
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           f"--nproc-per-node={nproc}", "--master-addr", "127.0.0.1",
           "--master-port", str(port), os.path.join(HERE, "bench.py"),
           "--gpus", str(nproc), "--steps", str(steps),
           "--warmup", str(warmup), "--batch-size", str(bs),
           "--json-out", out] + extra
    env = dict(os.environ, **env_extra)
    r = subprocess.run(cmd, capture_output=True, text=True, timeout=1800,
                       env=env)
    if r.returncode != 0:
        print(r.stdout[-2000:], r.stderr[-2000:], file=sys.stderr)
        raise RuntimeError(f"bench failed: {extra} {env_extra}")
    with open(out) as f:
        return json.loads(f.read())


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--nproc", type=int, default=4,
                    help="ranks; each rank is its own party here")
    ap.add_argument("--rates", type=str, default="1,1,1,0.1",
                    help="per-party uplink Gbit/s (one per rank)")
    ap.add_argument("--steps", type=int, default=8)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--batch-size", type=int, default=16)
    ap.add_argument("--image-size", type=int, default=28)
    ap.add_argument("--port", type=int, default=29660)
    ap.add_argument("--json-out", type=str, default=None)
    args = ap.parse_args()

    assert len(args.rates.split(",")) == args.nproc, \
        "--rates needs one value per rank"
    common = ["--image-size", str(args.image_size),
              "--parties", str(args.nproc), "--mode", "hips",
              "--party-wan-gbps", args.rates]
    configs = [
        ("collectives (all_reduce tier)", {"ENABLE_INTER_TS": "0"}, []),
        ("TSEngine relay (ENABLE_INTER_TS=1)", {"ENABLE_INTER_TS": "1"},
         []),
        ("TSEngine relay + fp16 wire", {"ENABLE_INTER_TS": "1"},
         ["--compress", "fp16"]),
    ]
    rows = []
    port = args.port
    for name, env_extra, extra in configs:
        res = run_bench(args.nproc, common + extra, args.steps,
                        args.warmup, args.batch_size, port, env_extra)
        rows.append((name, res))
        port += 2

    base = rows[0][1]["ms_per_step"]
    print(f"\nheterogeneous WAN {args.rates} Gbit/s, "
          f"{args.nproc} parties, geomx_cnn {args.image_size}px\n")
    print("| config | ms/step | speedup vs collectives |")
    print("|---|---|---|")
    out_rows = []
    for name, res in rows:
        sp = base / res["ms_per_step"]
        print(f"| {name} | {res['ms_per_step']:.1f} | {sp:.2f}x |")
        out_rows.append({"config": name, "ms_per_step": res["ms_per_step"],
                         "speedup": sp, "rates": args.rates})
    if args.json_out:
        with open(args.json_out, "w") as f:
            json.dump(out_rows, f, indent=1)


if __name__ == "__main__":
    main()
