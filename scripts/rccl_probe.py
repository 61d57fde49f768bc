#!/usr/bin/env python3
"""Probe: can N processes share ONE MI355X under RCCL?

If RCCL accepts duplicate devices in a communicator, every multi-rank
code path (sub-group all_gather, leader send/recv, bf16 wire) can be
exercised with real RCCL on a 1-GPU box. If not, the HiPS-vs-flat
matrix falls back to gloo transport with CUDA compute (scripts/
hips_vs_flat.py --device cuda).

Launch:  python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
           --master-addr 127.0.0.1 scripts/rccl_probe.py
"""
import os
import sys

import torch
import torch.distributed as dist


def main():
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    torch.cuda.set_device(0)  # every rank on the SAME GPU on purpose
    try:
        dist.init_process_group("nccl")
        t = torch.ones(1024, device="cuda") * (rank + 1)
        dist.all_reduce(t)
        torch.cuda.synchronize()
        expect = world * (world + 1) / 2
        ok = bool(torch.all(t == expect))
        # sub-group + p2p, the HiPS leader-tier primitives
        g = dist.new_group(ranks=list(range(world)))
        lst = [torch.empty_like(t) for _ in range(world)]
        dist.all_gather(lst, t, group=g)
        if world >= 2:
            if rank == 0:
                dist.send(t, dst=1)
            elif rank == 1:
                dist.recv(t, src=0)
        torch.cuda.synchronize()
        if rank == 0:
            print(f"RCCL_PROBE_OK allreduce={ok}")
    except Exception as e:  # noqa: BLE001
        print(f"RCCL_PROBE_FAIL rank={rank}: {type(e).__name__}: {e}",
              flush=True)
        sys.exit(1)


if __name__ == "__main__":
    main()
