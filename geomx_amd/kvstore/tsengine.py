"""TSEngine: throughput-aware overlay scheduling for the WAN tier.

The reference replaces the star-topology push/pull incast with a
dynamically scheduled relay tree: a central scheduler keeps a
throughput matrix ``A[i][j]`` (latest measured sender->receiver
throughput, -1 = never measured) and a per-round busy vector ``B``;
each node holding fresh data asks the scheduler for a receiver, the
scheduler answers epsilon-greedily (greedy once the row is fully
known, capped at MAX_GREED_RATE_TS = 0.9; pure exploration while any
candidate link is unmeasured), the node relays the data and reports
the achieved throughput in its next ask
(van.cc:1312-1458 ProcessAskPull{,Global}Command,
kv_app.h:1040-1076 AutoPull throughput measurement,
kvstore_dist.h:91-173 / kvstore_dist_server.h:228-310 WorkersMerge).

MI355X-native mapping: no scheduler process and no ASK/REPLY control
round-trips. All leaders hold an identical copy of A (synchronized by
a tiny all_gather of each leader's own row after every exchange — the
control plane rides the fast fabric) and derive the SAME schedule from
a shared seeded RNG, so every hop is a plain ``dist.send``/``recv``
pair over the leader group with zero coordination messages. Data hops
are paced/charged through the sender's WAN token bucket, so the
scheduler learns real (emulated) per-link throughput.

Two primitives mirror the two reference directions:

* ``merge(x)``   — push direction (WorkersMerge chain): holders pair up
  each round, senders transmit their partial sum to a scheduler-chosen
  receiver which accumulates; log2(P) rounds end with one root holding
  the global sum.
* ``spread(x, src)`` — pull direction (AutoPull relay tree): each
  round every holder forwards to a scheduler-chosen receiver until all
  leaders have the value.

``allreduce_sum`` = merge + spread-from-root: 2(P-1) point-to-point
messages total, each of which crosses ONE party boundary — no single
link ever carries the (P-1)-fold incast of a star exchange.
"""

from __future__ import annotations

import random
import time
from typing import Callable, List, Optional, Tuple

import torch
import torch.distributed as dist

from .. import comm

from .wan import TokenBucket

MAX_GREED_RATE_TS = 0.9  # reference: ps-lite van.h max_greed_rate


class TSScheduler:
    """The receiver-selection policy (van.cc:1336-1375), shared-state
    edition: every leader holds the same A and runs the same picks."""

    def __init__(self, num_nodes: int):
        self.n = num_nodes
        # latest measured throughput, bytes/s; -1 = unknown (reference
        # initializes A to -1 and overwrites with each report)
        self.A = [[-1.0] * num_nodes for _ in range(num_nodes)]

    def pick(self, sender: int, candidates: List[int],
             rng: random.Random) -> int:
        """Choose a receiver for `sender` among idle `candidates`.

        Greed rate: the reference computes
        ``num_known / (num_known + num_unknown)`` in INTEGER division,
        so it explores randomly until every candidate link has been
        measured and only then goes greedy at MAX_GREED_RATE_TS. We
        keep that (sensible) observed behavior explicitly.
        """
        if not candidates:
            raise ValueError("no idle candidates")
        row = self.A[sender]
        known = [c for c in candidates if row[c] >= 0]
        if len(known) == len(candidates) and known \
                and rng.random() < MAX_GREED_RATE_TS:
            return max(known, key=lambda c: row[c])
        return rng.choice(candidates)

    def update(self, sender: int, receiver: int, throughput: float):
        self.A[sender][receiver] = float(throughput)


class TSExchange:
    """Relay-tree collectives over a torch.distributed group.

    Parameters
    ----------
    group : ProcessGroup over the party leaders
    my_party : this rank's index within `ranks`
    ranks : global ranks of the leaders, indexed by party id
    wan : sender-side token bucket charged per data hop
    link_model : optional fn(src_party, dst_party, nbytes) -> seconds of
        extra emulated latency — lets tests build heterogeneous WANs

    Each primitive takes a `wire_dtype` (fp16 wire halves traffic;
    accumulation stays fp32, matching the fp32-master design).
    """

    def __init__(self, group, my_party: int, ranks: List[int],
                 wan: Optional[TokenBucket] = None,
                 link_model: Optional[Callable[[int, int, int],
                                               float]] = None):
        self.group = group
        self.me = my_party
        self.ranks = list(ranks)
        self.P = len(ranks)
        self.wan = wan
        self.link_model = link_model
        self.sched = TSScheduler(self.P)
        self._seq = 0
        # measurements this rank made since the last row sync
        self._my_row = [-1.0] * self.P

    # -- data hops ----------------------------------------------------
    def _send(self, x: torch.Tensor, dst_party: int,
              wire_dtype: torch.dtype):
        wire = x.to(wire_dtype)
        nbytes = wire.numel() * wire.element_size()
        t0 = time.perf_counter()
        if self.wan is not None:
            self.wan.charge(nbytes, sync_device=False)
        if self.link_model is not None:
            extra = self.link_model(self.me, dst_party, nbytes)
            if extra > 0:
                time.sleep(extra)
        comm.send(wire, dst=self.ranks[dst_party], group=self.group)
        if wire.is_cuda:
            # NCCL send only enqueues on the comm stream; sync so the
            # measured interval covers the actual transfer (on the
            # emulated WAN the token-bucket wait above dominates either
            # way, so A reflects the configured per-party rates)
            torch.cuda.synchronize()
        dt = max(time.perf_counter() - t0, 1e-9)
        tput = nbytes / dt
        self._my_row[dst_party] = tput
        self.sched.update(self.me, dst_party, tput)

    def _recv(self, numel: int, src_party: int, device,
              wire_dtype: torch.dtype) -> torch.Tensor:
        wire = torch.empty(numel, dtype=wire_dtype, device=device)
        comm.recv(wire, src=self.ranks[src_party], group=self.group)
        return wire.float()

    def _sync_rows(self, device):
        """all_gather each leader's measured row so every copy of A is
        identical before the next schedule is drawn (the reference
        piggybacks reports on the next ASK; our control plane is one
        tiny collective on the fast fabric, never WAN-charged)."""
        row = torch.tensor(self._my_row, dtype=torch.float64,
                           device=device)
        rows = [torch.empty_like(row) for _ in range(self.P)]
        comm.all_gather(rows, row, group=self.group)
        for i, r in enumerate(rows):
            for j, v in enumerate(r.tolist()):
                if v >= 0:
                    self.sched.update(i, j, v)

    # -- schedules (identical on every rank: shared RNG + shared A) ---
    def _spread_schedule(self, src: int,
                         rng: random.Random) -> List[List[Tuple[int, int]]]:
        have = [src]
        lack = [p for p in range(self.P) if p != src]
        rounds = []
        while lack:
            sends = []
            avail = list(lack)
            for s in list(have):
                if not avail:
                    break
                r = self.sched.pick(s, avail, rng)
                avail.remove(r)
                sends.append((s, r))
            rounds.append(sends)
            have += [r for _, r in sends]
            lack = avail
        return rounds

    def _merge_schedule(self, rng: random.Random
                        ) -> Tuple[List[List[Tuple[int, int]]], int]:
        holders = list(range(self.P))
        rounds = []
        while len(holders) > 1:
            sends = []
            avail = list(holders)
            while len(avail) >= 2:
                s = avail.pop()            # deterministic sender choice
                r = self.sched.pick(s, avail, rng)
                avail.remove(r)
                sends.append((s, r))
            rounds.append(sends)
            holders = [r for _, r in sends] + avail
        return rounds, holders[0]

    # -- public primitives -------------------------------------------
    def spread(self, x: torch.Tensor, src: int,
               wire_dtype: torch.dtype = torch.float32) -> torch.Tensor:
        """Relay-broadcast `x` from party `src` to all leaders
        (DefaultAutoPull relay, kvstore_dist_server.h:1368)."""
        flat = x.reshape(-1).float()
        if wire_dtype != torch.float32:
            # round the root's own copy too: every replica must end
            # bit-identical, relays re-quantize idempotently
            flat = flat.to(wire_dtype).float()
        rng = random.Random(self._seq)
        self._seq += 1
        for sends in self._spread_schedule(src, rng):
            for s, r in sends:
                if self.me == s:
                    self._send(flat, r, wire_dtype)
                elif self.me == r:
                    flat = self._recv(flat.numel(), s, flat.device,
                                      wire_dtype)
        self._sync_rows(flat.device)
        return flat.reshape(x.shape)

    def merge(self, x: torch.Tensor,
              wire_dtype: torch.dtype = torch.float32
              ) -> Tuple[int, torch.Tensor]:
        """Relay-merge: sums every party's `x`; returns (root_party,
        sum) — the sum is valid only on the root (WorkersMerge chain)."""
        acc = x.reshape(-1).float().clone()
        rng = random.Random(self._seq)
        self._seq += 1
        rounds, root = self._merge_schedule(rng)
        for sends in rounds:
            for s, r in sends:
                if self.me == s:
                    self._send(acc, r, wire_dtype)
                elif self.me == r:
                    acc = acc + self._recv(acc.numel(), s, acc.device,
                                           wire_dtype)
        self._sync_rows(acc.device)
        return root, acc.reshape(x.shape)

    def stats(self) -> dict:
        """Observability: the learned throughput matrix (bytes/s, -1 =
        unmeasured) and exchange count — the data a scheduler operator
        would watch on a real WAN."""
        return {"A_bytes_per_s": [row[:] for row in self.sched.A],
                "exchanges": self._seq}

    def allreduce_sum(self, x: torch.Tensor,
                      wire_dtype: torch.dtype = torch.float32
                      ) -> torch.Tensor:
        root, total = self.merge(x, wire_dtype)
        return self.spread(total, root, wire_dtype)
