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

Message-loss tolerance (ps-lite Resender / PS_DROP_MSG parity,
src/resender.h + van.cc:871-877): every push payload is tagged with its
sequence number; `GEOMX_DROP_MSG=<pct>` makes the producer "lose" that
fraction of transmissions (fault injection). The server validates the
tag before consuming — a lost or stale slot stalls that party's stream
(never corrupts it) — and the producer retransmits un-ACKed pushes
after `GEOMX_RESEND_TIMEOUT_MS` (checked on every push/pull/flush). If
a gap can never be repaired (the producer died with payloads in
flight), the server skips it after `skip_timeout_s` and counts it in
`self.lost` — the reference's lossy-channel stance: a lost gradient
delays nobody.
"""

from __future__ import annotations

import os
import random
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

    def __init__(self, store, topo, device, wan=None, poll_s: float = 0.001,
                 drop_pct: Optional[float] = None,
                 resend_timeout_s: Optional[float] = None,
                 skip_timeout_s: float = 5.0):
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
        # Resender / fault injection (PS_DROP_MSG is a percentage)
        if drop_pct is None:
            drop_pct = float(os.environ.get("GEOMX_DROP_MSG", "0")) / 100.0
        if resend_timeout_s is None:
            resend_timeout_s = float(
                os.environ.get("GEOMX_RESEND_TIMEOUT_MS", "50")) / 1000.0
        self.drop_pct = drop_pct
        self.resend_timeout_s = resend_timeout_s
        self.skip_timeout_s = skip_timeout_s
        self.lost = 0                 # server: pushes skipped as unrepairable
        self._rng = random.Random(0xA5 + getattr(topo, "rank", 0))
        # producer: (key, seq) -> [payload, last_tx_time]
        self._unacked: Dict[tuple, list] = {}
        self._stall: Dict[tuple, float] = {}  # server: (key,party) stall t0

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
            # keep the resender alive while blocked: the window can be
            # full precisely BECAUSE an earlier transmission was lost
            self._resend_unacked()
            time.sleep(self.poll_s)
        payload = seq.to_bytes(8, "little") + _tensor_bytes(party_sum)
        if self.wan is not None:
            self.wan.charge(len(payload), sync_device=False)
        self._unacked[(key, seq)] = [payload, 0.0]
        self._transmit(key, seq)
        self._resend_unacked()

    def _transmit(self, key, seq):
        """One transmission attempt, subject to injected loss. The seq
        announcement always carries the producer's HIGH-WATER mark so a
        retransmission of an old slot never regresses the counter."""
        pid = self.topo.party_id
        rec = self._unacked.get((key, seq))
        if rec is None:
            return
        rec[1] = time.time()
        if self.drop_pct > 0 and self._rng.random() < self.drop_pct:
            return  # "lost on the WAN" — announcement and data both
        self.store.set(self._k(key, f"p{pid}|d{seq % DEPTH}"), rec[0])
        self.store.set(self._k(key, f"p{pid}|seq"),
                       str(self._push_seq[key]).encode())

    def _resend_unacked(self):
        """Resender (src/resender.h): retransmit pushes un-ACKed after
        the timeout. Called on every push/pull/flush."""
        if not self._unacked:
            return
        pid = self.topo.party_id
        now = time.time()
        acks: Dict[object, int] = {}
        for (key, seq) in list(self._unacked):
            ack = acks.get(key)
            if ack is None:
                ack = acks[key] = self._get_int(self._k(key, f"p{pid}|ack"))
            if seq <= ack:
                del self._unacked[(key, seq)]
            elif now - self._unacked[(key, seq)][1] > self.resend_timeout_s:
                self._transmit(key, seq)

    def flush(self, timeout_s: float = 30.0) -> bool:
        """Producer: block until every push of this party is ACKed
        (resending as needed)."""
        end = time.time() + timeout_s
        while self._unacked and time.time() < end:
            self._resend_unacked()
            if self._unacked:
                time.sleep(self.poll_s)
        return not self._unacked

    def pull(self, key, timeout_s: float = 60.0) -> torch.Tensor:
        self._resend_unacked()
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
                        nxt = seen + 1
                        slot = self._k(key, f"p{p}|d{nxt % DEPTH}")
                        ok = False
                        if self.store.check([slot]):
                            payload = self.store.get(slot)
                            tag = int.from_bytes(payload[:8], "little")
                            ok = tag == nxt
                        if not ok:
                            # lost or stale slot: wait for the resender;
                            # give up only on an unrepairable gap (dead
                            # producer) and count the loss
                            t0 = self._stall.setdefault((key, p), time.time())
                            if time.time() - t0 > self.skip_timeout_s:
                                self.lost += 1
                            else:
                                break
                        self._stall.pop((key, p), None)
                        seen = nxt
                        if ok:
                            grad = _bytes_tensor(payload[8:], n, self.device)
                            st = self._stored[key]
                            if self.optimizer is not None:
                                self.optimizer.update(key, st, grad)
                            else:
                                st.add_(grad)
                            self.applied += 1
                        self._seen[(key, p)] = seen
                        self.store.set(self._k(key, f"p{p}|ack"),
                                       str(seen).encode())
                        if ok:
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
