#!/usr/bin/env python3
"""Microbenchmark of the geops HIP kernels: effective HBM bandwidth vs
the ~6.3 TB/s MI355X ceiling. Run on a GPU box:

    python scripts/kernel_bench.py [--n 67108864] [--iters 50]

Prints one line per kernel: name, bytes moved per call, ms, GB/s.
"""

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from geomx_amd import ops  # noqa: E402
from geomx_amd.ops import reference as ref  # noqa: E402


def timeit(fn, iters, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--n", type=int, default=1 << 26)  # 64M fp32 = 256 MB
    ap.add_argument("--iters", type=int, default=50)
    ap.add_argument("--json-out", type=str, default=None)
    args = ap.parse_args()
    assert torch.cuda.is_available()
    assert ops.native_available(), ops.native_error()
    from geomx_amd.ops import _geops as geops

    n = args.n
    dev = "cuda:0"
    g = torch.randn(n, device=dev)
    results = []

    def report(name, nbytes, sec):
        gbs = nbytes / sec / 1e9
        line = {"kernel": name, "bytes": nbytes, "ms": round(sec * 1e3, 4),
                "GBps": round(gbs, 1)}
        results.append(line)
        print(f"{name:24s} {nbytes/1e6:10.1f} MB {sec*1e3:8.3f} ms "
              f"{gbs:8.1f} GB/s")

    # --- 2bit quantize: read g + r, write r + n/16 words
    r = torch.zeros(n, device=dev)
    packed = torch.empty((n + 15) // 16, dtype=torch.int32, device=dev)
    sec = timeit(lambda: geops.quantize_2bit(g, r, packed, 0.5), args.iters)
    report("quantize_2bit", n * 4 * 3 + n // 4, sec)

    out = torch.empty(n, device=dev)
    sec = timeit(lambda: geops.dequantize_2bit(packed, out, 0.5), args.iters)
    report("dequantize_2bit", n * 4 + n // 4, sec)

    # --- bsc momentum: read g,u,v write u,v
    u = torch.zeros(n, device=dev)
    v = torch.zeros(n, device=dev)
    sec = timeit(lambda: geops.bsc_momentum(g, u, v, 0.9), args.iters)
    report("bsc_momentum", n * 4 * 5, sec)

    # --- bsc pack (1% capacity): read v (+u,v writes at 1%)
    k = int(n * 0.01)
    vals = torch.empty(k, device=dev)
    idx = torch.empty(k, dtype=torch.int32, device=dev)
    v.copy_(g)
    boundary = 2.3  # ~1% of a standard normal
    sec = timeit(lambda: geops.bsc_pack(v, u, vals, idx, boundary, -65530.0),
                 args.iters)
    report("bsc_pack(1%)", n * 4 * 2, sec)  # count pass + pack pass reads

    sec = timeit(lambda: geops.bsc_unpack(vals, idx, out, False), args.iters)
    report("bsc_unpack(1%)", n * 4 + k * 8, sec)

    # --- fused single-pass compress (momentum+count+pack), the
    # production bsc path: read g,u,v + write u,v in ONE sweep
    bt = torch.full((1,), 2.3, device=dev)
    sec = timeit(lambda: geops.bsc_compress_fused(g, u, v, vals, idx, bt,
                                                  0.9, -65530.0),
                 args.iters)
    report("bsc_fused(1%)", n * 4 * 5, sec)

    import geomx_amd.ops as _ops
    sec = timeit(lambda: _ops.bsc_compress(g, u, v, 0.01), args.iters)
    report("bsc_compress e2e", n * 4 * 5, sec)

    sec = timeit(lambda: geops.bsc_pull_pack(out, vals, idx, -65530.0),
                 args.iters)
    report("bsc_pull_pack", n * 4 * 2, sec)

    # --- dgt contribution
    nchunks = (n + 1023) // 1024
    contrib = torch.empty(nchunks, device=dev)
    sec = timeit(lambda: geops.dgt_contribution(g, contrib, 1024), args.iters)
    report("dgt_contribution", n * 4, sec)

    # --- 4bit quantize (minmax pass + pack pass)
    p4 = torch.empty((n + 1) // 2, dtype=torch.uint8, device=dev)
    mm = torch.empty(nchunks * 2, device=dev).reshape(nchunks, 2)
    empty = torch.Tensor()
    sec = timeit(lambda: geops.quantize_4bit(g, empty, p4, mm, 1024),
                 args.iters)
    report("quantize_4bit", n * 4 * 2 + n // 2, sec)

    sec = timeit(lambda: geops.dequantize_4bit(p4, mm, out, 1024), args.iters)
    report("dequantize_4bit", n // 2 + n * 4, sec)

    # --- optimizers
    w = torch.randn(n, device=dev)
    sec = timeit(lambda: geops.sgd_update(w, g, 0.01, 0.0, 1.0), args.iters)
    report("sgd_update", n * 4 * 3, sec)

    m = torch.zeros(n, device=dev)
    sec = timeit(lambda: geops.sgd_mom_update(w, g, m, 0.01, 0.9, 0.0, 1.0),
                 args.iters)
    report("sgd_mom_update", n * 4 * 5, sec)

    vv = torch.zeros(n, device=dev)
    sec = timeit(lambda: geops.adam_update(w, g, m, vv, 10, 0.01, 0.9, 0.999,
                                           1e-8, 0.0, 1.0), args.iters)
    report("adam_update", n * 4 * 7, sec)

    pw = w.clone()
    sec = timeit(lambda: geops.dcasgd_update(w, g, pw, empty, 0.01, 0.04, 0.0,
                                             0.0, 1.0), args.iters)
    report("dcasgd_update", n * 4 * 5, sec)

    if args.json_out:
        with open(args.json_out, "w") as f:
            json.dump({"n": n, "iters": args.iters, "results": results}, f,
                      indent=1)


if __name__ == "__main__":
    main()
