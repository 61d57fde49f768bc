"""Helpers for multi-process CPU (gloo) tests — the `local.sh` pattern
from the reference's ps-lite tests (fork N local processes differing
only by role env), rebuilt on torch.multiprocessing."""

import os
import traceback

import torch.multiprocessing as mp


def _child(rank, world_size, port, fn, args, err_queue):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        os.environ["RANK"] = str(rank)
        os.environ["WORLD_SIZE"] = str(world_size)
        os.environ["LOCAL_RANK"] = str(rank)
        import torch.distributed as dist
        dist.init_process_group("gloo", rank=rank, world_size=world_size)
        try:
            fn(rank, world_size, *args)
        finally:
            dist.destroy_process_group()
    except Exception:
        err_queue.put((rank, traceback.format_exc()))
        raise


def run_dist(world_size, fn, *args, timeout=120):
    """Run fn(rank, world_size, *args) in world_size processes over gloo."""
    from conftest import free_port
    port = free_port()
    ctx = mp.get_context("spawn")
    err_queue = ctx.SimpleQueue()
    procs = []
    for r in range(world_size):
        p = ctx.Process(target=_child,
                        args=(r, world_size, port, fn, args, err_queue))
        p.start()
        procs.append(p)
    failed = []
    for r, p in enumerate(procs):
        p.join(timeout)
        if p.is_alive():
            p.terminate()
            p.join(10)
            failed.append((r, "timeout"))
        elif p.exitcode != 0:
            failed.append((r, f"exit {p.exitcode}"))
    msgs = []
    while not err_queue.empty():
        rank, tb = err_queue.get()
        msgs.append(f"--- rank {rank} ---\n{tb}")
    if failed or msgs:
        raise AssertionError(
            f"dist test failed: {failed}\n" + "\n".join(msgs))
