"""Failure detection: heartbeat registry over the torch.distributed
TCPStore.

Parity target: ps-lite heartbeats — nodes send HEARTBEAT to the
scheduler every PS_HEARTBEAT_INTERVAL seconds; the scheduler records
timestamps and the app can query dead nodes
(van.cc:1128-1140; Postoffice::UpdateHeartbeat/GetDeadNodes,
postoffice.h:179-187; surfaced as get_num_dead_node,
kvstore_dist.h:225-234).

MI355X-native: there is no separate scheduler process; every rank
writes `hb_<rank>` = monotonic timestamp into the rendezvous TCPStore
from a daemon thread, and any rank can count peers whose heartbeat is
older than `timeout` seconds.
"""

from __future__ import annotations

import threading
import time
from typing import List, Optional

import torch.distributed as dist


class HeartbeatMonitor:
    def __init__(self, interval_s: float = 1.0, store=None,
                 rank: Optional[int] = None,
                 world_size: Optional[int] = None):
        if store is None:
            if not dist.is_initialized():
                raise RuntimeError("HeartbeatMonitor needs an initialized "
                                   "process group or an explicit store")
            store = dist.distributed_c10d._get_default_store()
            rank = dist.get_rank() if rank is None else rank
            world_size = dist.get_world_size() if world_size is None else world_size
        self.store = store
        self.rank = rank
        self.world_size = world_size
        self.interval = interval_s
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self._beat()  # initial heartbeat so peers see us immediately

    def _key(self, r: int) -> str:
        return f"geomx_hb_{r}"

    def _beat(self):
        self.store.set(self._key(self.rank), repr(time.time()))

    def start(self):
        if self._thread is not None:
            return

        def loop():
            while not self._stop.wait(self.interval):
                self._beat()

        self._thread = threading.Thread(target=loop, daemon=True)
        self._thread.start()

    def stop(self):
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=5)
            self._thread = None

    def last_heartbeat(self, rank: int) -> Optional[float]:
        try:
            v = self.store.get(self._key(rank))
        except Exception:
            return None
        return float(v.decode() if isinstance(v, bytes) else v)

    def dead_nodes(self, timeout_s: float) -> List[int]:
        """Ranks whose last heartbeat is older than timeout_s (the
        GetDeadNodes analog)."""
        now = time.time()
        dead = []
        for r in range(self.world_size):
            hb = self.last_heartbeat(r)
            if hb is None or (now - hb) > timeout_s:
                dead.append(r)
        return dead

    def get_num_dead_node(self, timeout_s: float = 60.0) -> int:
        return len(self.dead_nodes(timeout_s))
