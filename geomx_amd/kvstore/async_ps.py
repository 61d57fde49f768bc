"""True-async global tier: a store-based parameter server.

The lockstep dist_async mode reproduces MixedSync's UPDATE MATH
(sequential per-party optimizer steps) but not its CADENCE — with
collectives, a slow party still stalls the fast ones. This module
implements the reference's actual async behavior
(DataHandleAsyncDefault, kvstore_dist_server.h:1519-1611): each party's
leader pushes its aggregated gradient whenever it is ready, the global
server applies each push ON ARRIVAL, and pulls return the latest
published parameters without waiting for anyone.

Transport: the torch.distributed rendezvous TCPStore (already shared by
every rank). Keys are double-buffered so the key set stays bounded:

  aps|{key}|p{party}|seq          -> push sequence counter (int)
  aps|{key}|p{party}|ack          -> server's consumed counter (int)
  aps|{key}|p{party}|d{seq % D}   -> push payload ring, D = 8 deep
  aps|{key}|ver                   -> published parameter version (int)
  aps|{key}|v{ver % 2}            -> parameter payload (fp32 bytes)

The server consumes EVERY push in sequence (the reference's async
server applies each arriving push, not just the newest); the D-deep
payload ring gives producers bounded-buffer backpressure — a leader
more than D-1 pushes ahead of the server waits (with a timeout) before
overwriting an unconsumed slot.

The global server (leader of party 0) runs a daemon consumer thread
polling the push counters; everything is bounded by timeouts, so a dead
peer can never hang training (the heartbeat layer reports it instead).

This is intra-datacenter-synchronous / inter-datacenter-asynchronous —
exactly the reference's MixedSync split. The WAN token bucket still
prices every payload that crosses the party boundary.
"""

from __future__ import annotations

import threading
import time
from typing import Dict, Optional

import numpy as np
import torch


def _tensor_bytes(t: torch.Tensor) -> bytes:
    return t.detach().float().cpu().numpy().tobytes()


def _bytes_tensor(b: bytes, numel: int, device) -> torch.Tensor:
    arr = np.frombuffer(bytearray(b), dtype=np.float32, count=numel)
    return torch.from_numpy(arr).to(device)


DEPTH = 8  # payload ring depth per (key, party)


class AsyncPSGlobal:
    """Leader-side endpoint of the async global tier. The leader of
    party 0 additionally hosts the server (consumer thread + optimizer
    + authoritative parameters)."""

    def __init__(self, store, topo, device, wan=None, poll_s: float = 0.001):
        self.store = store
        self.topo = topo
        self.device = device
        self.wan = wan
        self.poll_s = poll_s
        self.is_server = topo.is_leader and topo.party_id == 0
        self.keys: Dict[object, int] = {}          # key -> numel
        self._push_seq: Dict[object, int] = {}     # my outgoing counters
        self._seen: Dict[tuple, int] = {}          # server: consumed counters
        self._stored: Dict[object, torch.Tensor] = {}  # server: params
        self.optimizer = None                      # server: ServerOptimizer
        self.applied = 0
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    # -- key helpers ----------------------------------------------------
    def _k(self, key, suffix):
        return f"aps|{key}|{suffix}"

    def _get_int(self, k, default=0):
        # store.add(k, 0) is the atomic read-or-create: a plain get()
        # BLOCKS (store timeout, minutes) while a key does not exist yet
        try:
            return int(self.store.add(k, 0))
        except Exception:
            return default

    # -- lifecycle ------------------------------------------------------
    def register(self, key, init_value: torch.Tensor):
        """Register a key. Elastic rejoin is implicit (ps-lite
        `is_recovery` ADD_NODE semantics, van.cc:372-394: a recovered
        node skips barriers and re-pulls live state): if the store
        already carries state for this key, a restarted leader resumes
        its push sequence where it left off, and a restarted server
        adopts the last published parameters and the per-party consumed
        counters instead of re-initializing. Optimizer state is not in
        the store — reload it via load_optimizer_states if it matters."""
        n = init_value.numel()
        self.keys[key] = n
        # resume the outgoing counter: the store's seq is authoritative
        # (0 for a fresh start). Without this a restarted leader would
        # emit seq 1 against a server that already consumed past it,
        # and every subsequent push would be silently ignored.
        self._push_seq[key] = self._get_int(
            self._k(key, f"p{self.topo.party_id}|seq"))
        if self.is_server:
            ver = self._get_int(self._k(key, "ver"))
            if ver >= 1:
                # server rejoin: published params + acks survive us
                payload = self.store.get(self._k(key, f"v{ver % 2}"))
                self._stored[key] = _bytes_tensor(payload, n, self.device)
                for p in range(self.topo.num_parties):
                    self._seen[(key, p)] = self._get_int(
                        self._k(key, f"p{p}|ack"))
                return
            flat = init_value.detach().reshape(-1).float().to(self.device)
            self._stored[key] = flat.clone()
            self._publish(key)
            # counters default to 0 when absent: no explicit reset (a
            # reset could race with another leader's first push). NOTE:
            # one store-transport kvstore per process group — counters
            # are namespaced by key only.
            for p in range(self.topo.num_parties):
                self._seen[(key, p)] = 0

    def start(self):
        if self.is_server and self._thread is None:
            self._thread = threading.Thread(target=self._serve, daemon=True)
            self._thread.start()

    def stop(self):
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=10)
            self._thread = None

    # -- worker (leader) side -------------------------------------------
    def push(self, key, party_sum: torch.Tensor, timeout_s: float = 60.0):
        seq = self._push_seq[key] + 1
        self._push_seq[key] = seq
        pid = self.topo.party_id
        # bounded-buffer backpressure: never overwrite unconsumed slots
        end = time.time() + timeout_s
        while seq - self._get_int(self._k(key, f"p{pid}|ack")) >= DEPTH:
            if time.time() > end:
                raise TimeoutError(
                    f"async PS server lagging > {DEPTH} pushes on {key!r}")
            time.sleep(self.poll_s)
        payload = _tensor_bytes(party_sum)
        if self.wan is not None:
            self.wan.charge(len(payload), sync_device=False)
        self.store.set(self._k(key, f"p{pid}|d{seq % DEPTH}"), payload)
        self.store.set(self._k(key, f"p{pid}|seq"), str(seq).encode())

    def pull(self, key, timeout_s: float = 60.0) -> torch.Tensor:
        ver = self._get_int(self._k(key, "ver"))
        end = time.time() + timeout_s
        while ver < 1:  # server has not published the initial params yet
            if time.time() > end:
                raise TimeoutError(f"async PS never published {key!r}")
            time.sleep(self.poll_s)
            ver = self._get_int(self._k(key, "ver"))
        payload = self.store.get(self._k(key, f"v{ver % 2}"))
        if self.wan is not None:
            self.wan.charge(len(payload), sync_device=False)
        return _bytes_tensor(payload, self.keys[key], self.device)

    # -- server side -----------------------------------------------------
    def _publish(self, key):
        st = self._stored[key]
        ver = self._get_int(self._k(key, "ver")) + 1
        self.store.set(self._k(key, f"v{ver % 2}"), _tensor_bytes(st))
        self.store.set(self._k(key, "ver"), str(ver).encode())

    def _serve(self):
        """Consume pushes on arrival; apply the optimizer (or plain
        accumulate) per push; publish updated params."""
        while not self._stop.is_set():
            progressed = False
            for key, n in list(self.keys.items()):
                for p in range(self.topo.num_parties):
                    seq = self._get_int(self._k(key, f"p{p}|seq"))
                    seen = self._seen.get((key, p), 0)
                    while seen < seq:
                        # consume EVERY push in order (the async server
                        # applies each arriving push)
                        seen += 1
                        payload = self.store.get(
                            self._k(key, f"p{p}|d{seen % DEPTH}"))
                        grad = _bytes_tensor(payload, n, self.device)
                        st = self._stored[key]
                        if self.optimizer is not None:
                            self.optimizer.update(key, st, grad)
                        else:
                            st.add_(grad)
                        self._seen[(key, p)] = seen
                        self.store.set(self._k(key, f"p{p}|ack"),
                                       str(seen).encode())
                        self.applied += 1
                        self._publish(key)
                        progressed = True
            if not progressed:
                time.sleep(self.poll_s)

    def drain(self, timeout_s: float = 30.0) -> bool:
        """Server: block until every announced push has been applied
        (testing/shutdown helper)."""
        end = time.time() + timeout_s
        while time.time() < end:
            done = True
            for key in self.keys:
                for p in range(self.topo.num_parties):
                    if self._get_int(self._k(key, f"p{p}|seq")) > \
                            self._seen.get((key, p), 0):
                        done = False
            if done:
                return True
            time.sleep(0.005)
        return False
