"""Profiling: torch.profiler front-end with GeoMX's profiler control
surface, including the geo-specific "worker remotely drives server-side
profilers" verb.

Parity targets:
  - python/mxnet/profiler.py:33-196 set_config/set_state/pause/resume/
    dump.
  - KVStoreServerProfilerCommand {kSetConfig,kState,kPause,kDump}
    (include/mxnet/kvstore.h:49) sent via SetServerProfilerCommand
    (kvstore_dist.h:200-205), handled server-side with filenames
    prefixed rank<N>_ (kvstore_dist_server.h:409-456).

Here the "server" is the leader rank; the command travels in-band as a
broadcast object list on the process group, and every rank writes
rank<N>_-prefixed chrome traces — same observable artifact layout.
"""

from __future__ import annotations

import os
from typing import Optional

import torch
import torch.distributed as dist


class Profiler:
    def __init__(self, filename: str = "profile.json",
                 with_stack: bool = False):
        self.filename = filename
        self.with_stack = with_stack
        self._prof: Optional[torch.profiler.profile] = None
        self._paused = False

    # -- mxnet-profiler-like control -----------------------------------
    def set_config(self, filename: str = None, with_stack: bool = None):
        if filename is not None:
            self.filename = filename
        if with_stack is not None:
            self.with_stack = with_stack

    _running = False

    def set_state(self, state: str):
        if state == "run":
            if not self._running:
                acts = [torch.profiler.ProfilerActivity.CPU]
                if torch.cuda.is_available():
                    acts.append(torch.profiler.ProfilerActivity.CUDA)
                self._prof = torch.profiler.profile(
                    activities=acts, with_stack=self.with_stack)
                self._prof.__enter__()
                self._running = True
        elif state == "stop":
            if self._prof is not None and self._running:
                self._prof.__exit__(None, None, None)
                self._running = False
        else:
            raise ValueError(state)

    def pause(self):
        self._paused = True

    def resume(self):
        self._paused = False

    def dump(self, rank: Optional[int] = None):
        if self._prof is None:
            return None
        if self._running:  # chrome export needs a stopped profiler
            self.set_state("stop")
        if rank is None:
            rank = dist.get_rank() if dist.is_initialized() else 0
        base = os.path.basename(self.filename)
        d = os.path.dirname(self.filename) or "."
        out = os.path.join(d, f"rank{rank}_{base}")
        self._prof.export_chrome_trace(out)
        return out


# global default profiler (mx.profiler module-level API parity)
_default = Profiler()
set_config = _default.set_config
set_state = _default.set_state
pause = _default.pause
resume = _default.resume
dump = _default.dump


class ServerProfilerCommand:
    """The kvstore command verbs a worker can send to drive SERVER-side
    profilers (KVStoreServerProfilerCommand). SPMD transport: rank 0
    broadcasts the verb; leader ranks act on it."""
    SET_CONFIG = 0
    STATE = 1
    PAUSE = 2
    DUMP = 3


def send_server_profiler_command(kvstore, command: int, params: str = ""):
    """Worker-side: broadcast a profiler command; leaders (the 'servers')
    execute it on their local default profiler."""
    obj = [int(command), str(params)]
    if dist.is_initialized() and kvstore.topo.world_size > 1:
        dist.broadcast_object_list(obj, src=0)
    cmd, payload = obj
    if kvstore.topo.is_leader:
        if cmd == ServerProfilerCommand.SET_CONFIG:
            _default.set_config(filename=payload or None)
        elif cmd == ServerProfilerCommand.STATE:
            _default.set_state(payload or "run")
        elif cmd == ServerProfilerCommand.PAUSE:
            _default.pause()
        elif cmd == ServerProfilerCommand.DUMP:
            return _default.dump(rank=kvstore.topo.rank)
    return None
