"""Collective wrappers that pair CUDA compute with a CPU (gloo) wire.

On an RCCL (nccl) backend these are pass-throughs. On gloo, CUDA
tensors are staged through pinned CPU copies around the collective so
multi-process runs can keep COMPUTE on the GPU while the wire runs over
gloo — the configuration used to measure the HiPS-vs-flat matrix on a
single MI355X (RCCL rejects two ranks on one device; scripts/
rccl_probe.py). The byte counts on the wire are the real payloads, so
compression ratios measure truthfully; only the transport medium
differs from the 8-GPU RCCL path, which uses the same call sites.
"""

from __future__ import annotations

from typing import List, Optional

import torch
import torch.distributed as dist


def _needs_staging(t: torch.Tensor, group) -> bool:
    if not t.is_cuda or not dist.is_initialized():
        return False
    try:
        return dist.get_backend(group) == "gloo"
    except Exception:  # noqa: BLE001 - group without backend info
        return dist.get_backend() == "gloo"


class _StagedWork:
    """Wraps an async gloo collective on a CPU staging buffer; wait()
    copies the result back into the CUDA tensor."""

    def __init__(self, work, cpu: torch.Tensor, out: torch.Tensor):
        self.work = work
        self.cpu = cpu
        self.out = out

    def wait(self):
        if self.work is not None:
            self.work.wait()
        self.out.copy_(self.cpu, non_blocking=False)
        return True


def all_reduce(t: torch.Tensor, op=None, group=None, async_op: bool = False):
    op = op if op is not None else dist.ReduceOp.SUM
    if not _needs_staging(t, group):
        return dist.all_reduce(t, op=op, group=group, async_op=async_op)
    cpu = t.detach().cpu()
    if async_op:
        work = dist.all_reduce(cpu, op=op, group=group, async_op=True)
        return _StagedWork(work, cpu, t)
    dist.all_reduce(cpu, op=op, group=group)
    t.copy_(cpu)
    return None


def reduce(t: torch.Tensor, dst: int, op=None, group=None,
           async_op: bool = False):
    op = op if op is not None else dist.ReduceOp.SUM
    if not _needs_staging(t, group):
        return dist.reduce(t, dst=dst, op=op, group=group, async_op=async_op)
    cpu = t.detach().cpu()
    if async_op:
        work = dist.reduce(cpu, dst=dst, op=op, group=group, async_op=True)
        return _StagedWork(work, cpu, t)
    dist.reduce(cpu, dst=dst, op=op, group=group)
    t.copy_(cpu)
    return None


def broadcast(t: torch.Tensor, src: int, group=None):
    if not _needs_staging(t, group):
        return dist.broadcast(t, src=src, group=group)
    cpu = t.detach().cpu()
    dist.broadcast(cpu, src=src, group=group)
    t.copy_(cpu)
    return None


def all_gather(out: List[torch.Tensor], t: torch.Tensor, group=None,
               async_op: bool = False):
    if not _needs_staging(t, group):
        return dist.all_gather(out, t, group=group, async_op=async_op)
    cpu_in = t.detach().cpu()
    cpu_out = [torch.empty_like(cpu_in) for _ in out]
    if async_op:
        work = dist.all_gather(cpu_out, cpu_in, group=group, async_op=True)

        class _GatherWork:
            def wait(self_inner):
                work.wait()
                for o, c in zip(out, cpu_out):
                    o.copy_(c)
                return True
        return _GatherWork()
    dist.all_gather(cpu_out, cpu_in, group=group)
    for o, c in zip(out, cpu_out):
        o.copy_(c)
    return None


def gather(t: torch.Tensor, gather_list, dst: int, group=None,
           async_op: bool = False):
    if not _needs_staging(t, group):
        return dist.gather(t, gather_list=gather_list, dst=dst, group=group,
                           async_op=async_op)
    cpu_in = t.detach().cpu()
    cpu_list = [torch.empty_like(cpu_in) for _ in gather_list] \
        if gather_list is not None else None
    if async_op:
        work = dist.gather(cpu_in, gather_list=cpu_list, dst=dst,
                           group=group, async_op=True)

        class _GWork:
            def wait(self_inner):
                work.wait()
                if gather_list is not None:
                    for o, c in zip(gather_list, cpu_list):
                        o.copy_(c)
                return True
        return _GWork()
    dist.gather(cpu_in, gather_list=cpu_list, dst=dst, group=group)
    if gather_list is not None:
        for o, c in zip(gather_list, cpu_list):
            o.copy_(c)
    return None


def send(t: torch.Tensor, dst: int, group=None):
    if not _needs_staging(t, group):
        return dist.send(t, dst=dst, group=group)
    dist.send(t.detach().cpu(), dst=dst, group=group)


def recv(t: torch.Tensor, src: Optional[int] = None, group=None):
    if not _needs_staging(t, group):
        return dist.recv(t, src=src, group=group)
    cpu = torch.empty(t.shape, dtype=t.dtype, device="cpu")
    dist.recv(cpu, src=src, group=group)
    t.copy_(cpu)
