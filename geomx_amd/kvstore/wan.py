"""WAN emulation: token-bucket bandwidth cap on inter-party traffic.

The reference runs its inter-datacenter plane over real WANs (ps-lite
global plane, van.cc:916+). On one 8-GPU MI355X node we emulate the WAN
by pacing every byte that crosses a party boundary through a token
bucket of configured rate, so Bi-Sparse / FP16 / MPQ / DGT traffic
reductions translate into measurable wall-clock speedups
(BASELINE.json: "injected bandwidth cap on the inter-group RCCL
communicator").

Charging model (documented so flat-vs-HiPS is apples-to-apples): for a
collective moving `nbytes` of payload among P parties, the bytes that
cross party boundaries per link are
  all_reduce : 2 * nbytes * (P-1)/P
  reduce     :     nbytes * (P-1)/P
  broadcast  :     nbytes * (P-1)/P
  gather     :     nbytes * (P-1)/P   (per-source payloads summed by caller)
A flat (non-hierarchical) all_reduce over the whole world is charged as
if each party boundary carried its ring share: same formula with its
full payload — which is exactly why HiPS + compression wins under a cap.
"""

from __future__ import annotations

import time


import torch


class TokenBucket:
    """Pace traffic to `gbps` gigabit/s, plus `rtt_ms` of propagation
    delay per transfer. Thread-unsafe by design (one per kvstore;
    kvstore calls are already serialized per rank).

    Each charge() models ONE transfer over the party's uplink: it
    completes after a full RTT (handshake + propagation — optimistic
    one-RTT for a pipelined collective, and naturally one RTT PER HOP
    for relay-tree exchanges, which is exactly the latency cost a deep
    overlay pays on a real WAN) plus the serialization time
    bytes/rate. rtt_ms=0 (default) keeps the bandwidth-only model."""

    def __init__(self, gbps: float, rtt_ms: float = 0.0):
        self.gbps = float(gbps)
        self.rtt_s = float(rtt_ms) / 1e3
        self._debt_until = 0.0  # monotonic time when the link is free again
        self.total_bytes = 0
        self.total_wait = 0.0

    @property
    def enabled(self) -> bool:
        return self.gbps > 0 or self.rtt_s > 0

    def _serialize_s(self, nbytes: float) -> float:
        if self.gbps <= 0:
            return 0.0
        return nbytes * 8.0 / (self.gbps * 1e9)

    def charge(self, nbytes: float, sync_device: bool = True):
        """Block until the emulated link would have finished moving nbytes."""
        if not self.enabled or nbytes <= 0:
            return
        if sync_device and torch.cuda.is_available():
            torch.cuda.synchronize()
        now = time.perf_counter()
        # propagation does not occupy the link; serialization does
        start = max(now, self._debt_until)
        self._debt_until = start + self._serialize_s(nbytes)
        self.total_bytes += nbytes
        done = self._debt_until + self.rtt_s
        wait = done - now
        if wait > 0:
            self.total_wait += wait
            _sleep_precise(wait)

    def charge_async(self, nbytes: float) -> float:
        """Reserve link time WITHOUT blocking; returns the monotonic time
        at which the emulated transfer completes. Used by the pipelined
        (one-step-stale) WAN tier: the link 'transfers' while compute
        proceeds; wait_until() at the apply point sleeps only the
        remainder."""
        if not self.enabled or nbytes <= 0:
            return 0.0
        now = time.perf_counter()
        start = max(now, self._debt_until)
        self._debt_until = start + self._serialize_s(nbytes)
        self.total_bytes += nbytes
        return self._debt_until + self.rtt_s

    def wait_until(self, ready_time: float):
        if ready_time <= 0:
            return
        now = time.perf_counter()
        if ready_time > now:
            self.total_wait += ready_time - now
            _sleep_precise(ready_time - now)

    def stats(self):
        return {"gbps": self.gbps, "total_bytes": self.total_bytes,
                "total_wait_s": self.total_wait}


def _sleep_precise(seconds: float):
    """time.sleep with a short spin tail for sub-ms accuracy."""
    end = time.perf_counter() + seconds
    if seconds > 0.002:
        time.sleep(seconds - 0.001)
    while time.perf_counter() < end:
        pass


def cross_party_bytes(op: str, nbytes: int, num_parties: int) -> float:
    """Bytes crossing the busiest party's WAN link for a collective whose
    per-party payload is `nbytes`. Incast/outcast collectives (gather /
    broadcast at one root) are charged at the root's link — (P-1)*payload
    — which is exactly the star-topology bottleneck GeoMX's TSEngine
    attacks; ring-friendly collectives amortize to (P-1)/P shares."""
    p = num_parties
    if p <= 1:
        return 0.0
    frac = (p - 1) / p
    if op == "all_reduce":
        return 2.0 * nbytes * frac
    if op in ("reduce", "reduce_scatter"):
        return nbytes * frac
    if op in ("gather", "broadcast", "scatter"):
        return nbytes * (p - 1)        # root in/out-cast bottleneck
    if op == "all_gather":
        return nbytes * (p - 1)        # ring relay: each link carries (P-1) shares
    if op == "send":
        return float(nbytes)
    raise ValueError(op)
