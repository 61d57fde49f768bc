"""KVStoreDist: hierarchical (HiPS) parameter server over RCCL/xGMI.

Re-expresses GeoMX's two-level server message flow
(src/kvstore/kvstore_dist_server.h:1213-1366 DataHandleSyncDefault;
src/kvstore/kvstore_dist.h:460-720 worker Push_/PullImpl) as nested
torch.distributed collectives:

  worker push  -> intra-party reduce(SUM) to the party leader
                  (= N workers' pushes aggregated by the local server)
  leader tier  -> inter-party exchange on the leader group, optionally
                  compressed (Bi-Sparse / FP16 / MPQ / DGT) and paced by
                  the WAN token bucket
                  (= local servers re-pushing sums to the global server)
  owner/global -> fused HIP optimizer update on the authoritative fp32
                  copy (= global server ApplyUpdates), or plain store of
                  the aggregated gradient when no optimizer is set
                  (update-on-worker mode, examples/cnn_bsc.py)
  worker pull  <- leader-group broadcast (optionally pull-compressed,
                  BSCPullCompress) + intra-party broadcast

Global-tier strategies:
  * "sharded"    — each key has an owner leader (round-robin = MultiGPS,
                   kvstore_dist_server.h:1770-1810): gather to owner,
                   update there, broadcast back on pull.
  * "replicated" — leaders all_gather the (compressed) party sums and
                   each replays the identical deterministic update, so
                   pull needs no WAN traffic. This is the xGMI-native
                   incast-free strategy (the role TSEngine's relay trees
                   play in the reference, van.cc:1312-1458).

Synchronization modes:
  * dist_sync  (FSA)       — both tiers synchronous.
  * dist_async (MixedSync) — the global tier applies each party's
    contribution as a SEPARATE sequential optimizer step (the async
    global server applies every leader push on arrival,
    kvstore_dist_server.h:1519-1611); DCASGD compensates staleness.
  * HFA — leaders store party aggregates locally and only every K2-th
    push syncs globally, transmitting the milestone DELTA
    (stored - milestone)/P (kvstore_dist_server.h:959-972,1324-1343).

SPMD ordering contract: the reference is message-passing, so workers
may push/pull keys in any order; this design rides collectives, so
every rank must issue the SAME sequence of key operations. push() is
exempt in effect — it defers its WAN tier, and the flush applies every
pending key in deterministic (priority, seq) order at the next
state-reading call, so ranks may interleave OTHER work freely between
pushes. pull() order must match across ranks whenever the pull has a
wire (sharded dense mode); in "replicated" mode (and for compressed
exchanges, where every leader already holds the result) pulls are
wire-free and order-independent (test_async_push_flush_order_ws2).
"""

from __future__ import annotations


import pickle
from typing import Dict, List, Optional

import torch
import torch.distributed as dist

from .. import comm, ops
from ..config import Config
from ..topology import Topology, init_topology
from .base import KVStoreBase
from .optimizer import OptimizerSpec, ServerOptimizer
from .wan import TokenBucket, cross_party_bytes


class _KeyState:
    __slots__ = ("shape", "numel", "dtype", "stored", "update_buf",
                 "bsc_u", "bsc_v", "residual_2bit", "milestone",
                 "push_count", "owner_party", "dgt_contrib", "dgt_residual",
                 "sliced", "padded", "residual_2bit_wan")

    def __init__(self, shape, numel, dtype):
        self.shape = shape
        self.numel = numel
        self.dtype = dtype
        self.stored: Optional[torch.Tensor] = None       # fp32 flat
        self.update_buf: Optional[torch.Tensor] = None   # fp32 flat (leaders)
        self.bsc_u: Optional[torch.Tensor] = None        # leader momentum corr
        self.bsc_v: Optional[torch.Tensor] = None        # leader error accum
        self.residual_2bit: Optional[torch.Tensor] = None  # worker residual
        self.milestone: Optional[torch.Tensor] = None    # HFA milestone (leader)
        self.dgt_contrib: Optional[torch.Tensor] = None  # EWMA chunk contribution
        self.dgt_residual: Optional[torch.Tensor] = None
        self.push_count = 0
        self.owner_party = 0
        self.sliced = False   # P3/MultiGPS big-tensor slicing across leaders
        self.padded = 0       # padded numel (multiple of P) when sliced
        self.residual_2bit_wan: Optional[torch.Tensor] = None  # leader WAN tier


class _UpdaterAdapter:
    """Wraps a python updater fn(key, grad, stored) in the
    ServerOptimizer interface (the reference's Updater + pickled-
    controller path, kvstore_server.py:30-100)."""

    def __init__(self, fn):
        self.fn = fn
        self.spec = None

    def update(self, key, w, grad, rescale=None):
        g = grad if rescale in (None, 1.0) else grad * rescale
        self.fn(key, g, w)


class KVStoreDist(KVStoreBase):
    def __init__(self, cfg: Config, topo: Optional[Topology] = None,
                 global_mode: str = "sharded"):
        cfg.validate()
        self.cfg = cfg
        self.topo = topo if topo is not None else init_topology(
            cfg.num_parties, cfg.party_sizes, cfg.backend, cfg.device)
        if global_mode not in ("sharded", "replicated"):
            raise ValueError(global_mode)
        self.global_mode = global_mode
        self.keys: Dict[object, _KeyState] = {}
        self._key_order: List[object] = []
        self.optimizer: Optional[ServerOptimizer] = None
        self.compression: Optional[Dict] = None
        self.wan = TokenBucket(cfg.wan_rate_for(self.topo.party_id),
                               rtt_ms=cfg.wan_rtt_ms)
        self._device = self.topo.device
        self._aps = None  # true-async global tier (async_transport=store)
        # TSEngine: throughput-scheduled relay tier replaces the
        # all_gather for dense/fp16 payloads in replicated dist_sync
        self._ts = None
        if (cfg.enable_ts and global_mode == "replicated"
                and self.topo.num_parties > 1 and self.topo.is_leader
                and cfg.mode == "dist_sync"):
            from .tsengine import TSExchange
            self._ts = TSExchange(self.topo.leader_group,
                                  self.topo.party_id,
                                  self.topo.leader_ranks, wan=self.wan)

    # ------------------------------------------------------------------
    # properties (GeoMX API parity: kvstore.py:501-565)
    # ------------------------------------------------------------------
    @property
    def type(self) -> str:
        # the reference reports the ORIGINAL type string (kvstore.cc:80
        # kv->type_ = tname), e.g. "dist_device_sync"
        return getattr(self, "_type_name", None) or self.cfg.mode

    @property
    def rank(self) -> int:
        return self.topo.rank

    @property
    def num_workers(self) -> int:
        return self.topo.num_workers

    @property
    def num_all_workers(self) -> int:
        return self.topo.num_all_workers

    @property
    def is_master_worker(self) -> bool:
        return self.topo.is_master_worker

    @property
    def num_parties(self) -> int:
        return self.topo.num_parties

    def barrier(self):
        self.flush()
        if dist.is_initialized():
            dist.barrier()

    _barrier = barrier

    def get_num_dead_node(self, node_id: int = -1, timeout_s: float = 60.0) -> int:
        """Heartbeat-based dead-peer count (kvstore_dist.h:225-234 parity;
        node_id is accepted for API compatibility, the count covers the
        whole world)."""
        if self.topo.world_size <= 1:
            return 0
        if not hasattr(self, "_hb"):
            from ..utils.health import HeartbeatMonitor
            self._hb = HeartbeatMonitor()
            self._hb.start()
        return self._hb.get_num_dead_node(timeout_s)

    # ------------------------------------------------------------------
    # configuration commands (the reference sends these as in-band server
    # commands: CommandType, kvstore_dist_server.h:49-52)
    # ------------------------------------------------------------------
    def set_optimizer(self, optimizer) -> None:
        if isinstance(optimizer, OptimizerSpec):
            spec = optimizer
        elif isinstance(optimizer, dict):
            spec = OptimizerSpec(**optimizer)
        else:
            raise TypeError("set_optimizer expects OptimizerSpec or dict")
        self.flush()  # pending pushes apply under the OLD optimizer
        # every rank creates it; only leaders/owners apply it (state is lazy)
        self.optimizer = ServerOptimizer(spec)
        if self._aps is not None:
            self._aps.optimizer = ServerOptimizer(spec)

    def set_updater(self, updater) -> None:
        """Install a custom python updater fn(key, grad, stored) that
        mutates `stored` in place (python/mxnet/kvstore.py:593-632
        parity). Replaces the fused-optimizer path for every key."""
        if not callable(updater):
            raise TypeError("updater must be callable")
        self.optimizer = _UpdaterAdapter(updater)
        if self._aps is not None:
            # the async store-transport server applies updates itself;
            # give it the same adapter (it only calls .update())
            self._aps.optimizer = self.optimizer

    # CommandType, kvstore_dist_server.h:49-52 (the C++ enum; the
    # reference's Python mirror kvstore.py:89-97 predates the inserted
    # kSyncGlobalMode and is off by one from 4 up — we follow the C++
    # server, which is what actually interprets the codes)
    CMD_CONTROLLER = 0
    CMD_SET_MULTI_PRECISION = 1
    CMD_STOP_SERVER = 2
    CMD_SYNC_MODE = 3
    CMD_SYNC_GLOBAL_MODE = 4
    CMD_SET_GRADIENT_COMPRESSION = 5
    CMD_SET_PROFILER_PARAMS = 6

    def _send_command_to_servers(self, head: int, body: str) -> None:
        """Generic in-band server command (kvstore.py:644-661 parity).
        SPMD transport: rank 0 broadcasts; leader ranks (the 'servers')
        interpret built-in CommandType codes, then any registered
        handler sees the command too (kController user payloads)."""
        self.flush()
        obj = [int(head), str(body)]
        if dist.is_initialized() and self.topo.world_size > 1:
            dist.broadcast_object_list(obj, src=0)
        if self.topo.is_leader:
            self._handle_builtin_command(obj[0], obj[1])
            if self._server_command_handler is not None:
                self._server_command_handler(obj[0], obj[1])

    def _handle_builtin_command(self, head: int, body: str) -> None:
        """Server-side CommandType dispatch (DataHandleEx command arm,
        kvstore_dist_server.h:320-345)."""
        if head == self.CMD_SYNC_MODE:
            # servers boot async and are switched to sync by command
            # (:329-330); body "0" flips back for symmetry
            self.cfg.mode = "dist_sync" if body != "0" else "dist_async"
        elif head == self.CMD_SYNC_GLOBAL_MODE:
            # global tier sync toggle (:332-333). Our lockstep global
            # tier shares cfg.mode; replaying the same flip keeps the
            # two commands equivalent here.
            self.cfg.mode = "dist_sync" if body != "0" else "dist_async"
        elif head == self.CMD_STOP_SERVER:
            self.close()
        elif head == self.CMD_SET_MULTI_PRECISION:
            # fp32 master copies are structural in this design (flat
            # fp32 buckets / fp32 stored values) — accept and record
            self._multi_precision = True
        elif head == self.CMD_SET_GRADIENT_COMPRESSION:
            import json as _json
            self.set_gradient_compression(_json.loads(body))
        elif head == self.CMD_SET_PROFILER_PARAMS:
            # body "<verb>:<params>" with the KVStoreServerProfilerCommand
            # verbs (set_config/state/pause/dump — kvstore_dist_server.h
            # :409-456); executes on this leader's default profiler
            from ..utils import profiler as _prof
            verb, _, payload = body.partition(":")
            v = int(verb)
            if v == _prof.ServerProfilerCommand.SET_CONFIG:
                _prof._default.set_config(filename=payload or None)
            elif v == _prof.ServerProfilerCommand.STATE:
                _prof._default.set_state(payload or "run")
            elif v == _prof.ServerProfilerCommand.PAUSE:
                _prof._default.pause()
            elif v == _prof.ServerProfilerCommand.DUMP:
                _prof._default.dump(rank=self.topo.rank)
        # kController carries user payloads; the registered handler
        # (below) is the consumer

    def set_learning_rate(self, lr: float) -> None:
        """Runtime LR update on the server optimizer (the reference
        re-sends the pickled optimizer via kController; here it is a
        direct call — state is preserved, unlike set_optimizer)."""
        if self.optimizer is None:
            raise RuntimeError("no optimizer set")
        self.optimizer.set_learning_rate(lr)
        if self._aps is not None and self._aps.optimizer is not None:
            self._aps.optimizer.set_learning_rate(lr)

    _server_command_handler = None
    _multi_precision = False

    def set_server_command_handler(self, fn) -> None:
        self._server_command_handler = fn

    def set_gradient_compression(self, compression_params: Dict) -> None:
        self.flush()  # pending pushes complete under the OLD scheme
        params = dict(compression_params)
        ctype = params.get("type")
        if ctype not in ("2bit", "bsc", "fp16", "mpq", "dgt", "bsc_dgt"):
            raise ValueError(f"unknown compression type {ctype!r}")
        if ctype == "2bit":
            params.setdefault("threshold", self.cfg.threshold)
            # untested/meaningless compositions are rejected loudly:
            # HFA exchanges PARAMETER deltas (not gradients) and the
            # store-transport async tier applies pushes on arrival —
            # neither carries a 2bit error-feedback residual contract.
            if self.cfg.use_hfa:
                raise ValueError("2bit compression is not supported with "
                                 "HFA (parameter-delta exchange)")
            if self.cfg.mode == "dist_async" \
                    and self.cfg.async_transport == "store":
                raise ValueError("2bit compression is not supported with "
                                 "async_transport='store'")
        if ctype in ("bsc", "mpq", "bsc_dgt"):
            params.setdefault("threshold", self.cfg.bsc_ratio)
            params.setdefault("size_lower_bound", self.cfg.size_lower_bound)
        self.compression = params

    # -- true-async global tier (async_transport="store") ---------------
    def _use_aps(self) -> bool:
        return (self.cfg.mode == "dist_async"
                and self.cfg.async_transport == "store"
                and self.topo.num_parties > 1
                and dist.is_initialized())

    def _ensure_aps(self):
        if self._aps is None:
            from .async_ps import AsyncPSGlobal
            store = dist.distributed_c10d._get_default_store()
            self._aps = AsyncPSGlobal(store, self.topo, self._device,
                                      wan=self.wan)
            if isinstance(self.optimizer, _UpdaterAdapter):
                # custom python updater: share the adapter (it is
                # stateless apart from the user's closure)
                self._aps.optimizer = self.optimizer
            elif self.optimizer is not None:
                self._aps.optimizer = ServerOptimizer(self.optimizer.spec)
        return self._aps

    def close(self):
        self.flush()
        if self._aps is not None:
            self._aps.flush(timeout_s=10)  # deliver any un-ACKed pushes
            self._aps.stop()

    # ------------------------------------------------------------------
    # init
    # ------------------------------------------------------------------
    def init(self, key, value) -> None:
        # list-of-keys form (kvstore.py:118: "a string or int, or a
        # list of them")
        if isinstance(key, (list, tuple)):
            for k, v in zip(key, value):
                self.init(k, v)
            return
        if key in self.keys:
            raise ValueError(f"key {key!r} already initialised")
        st = _KeyState(tuple(value.shape), value.numel(), value.dtype)
        st.owner_party = len(self._key_order) % self.topo.num_parties
        P = self.topo.num_parties
        if P > 1 and self.global_mode == "sharded" \
                and st.numel >= self.cfg.bigarray_bound:
            # P3/MultiGPS: slice the key uniformly across ALL leaders
            # (EncodeP3Key kvstore_dist.h:763-799; server-side sharding
            # kvstore_dist_server.h:1770-1810). Chunks are rounded to a
            # multiple of 64 elements so every leader's slice base stays
            # 256B-aligned — the fused-optimizer kernels require 16-byte
            # alignment (csrc/geops.cpp check_f32).
            st.sliced = True
            align = 64 * P
            st.padded = ((st.numel + align - 1) // align) * align
        self.keys[key] = st
        self._key_order.append(key)
        flat = value.detach().reshape(-1).float().to(self._device)
        # rank 0's value is authoritative at init (reference: first init wins)
        if dist.is_initialized() and self.topo.world_size > 1:
            comm.broadcast(flat, src=0)
        st.stored = flat.clone()
        if self.cfg.use_hfa:
            # the reference seeds the milestone with the initial params
            # (HandleHFAAccumulate first call, kvstore_dist_server.h:963
            # — the init pull response copies stored into the
            # milestone), so the first K2 exchange ships (params - w0)/P
            # rather than full params. Results are identical under the
            # synchronous exchange (milestone terms cancel), but the
            # delta magnitude — what compression sees — matches.
            st.milestone = flat.clone()
        if self._use_aps() and self.topo.is_leader:
            aps = self._ensure_aps()
            aps.register(key, st.stored)
            aps.start()

    def _state(self, key) -> _KeyState:
        st = self.keys.get(key)
        if st is None:
            raise KeyError(f"key {key!r} not initialised")
        return st

    # ------------------------------------------------------------------
    # push — ASYNC with priority (engine-var semantics of
    # kvstore_dist.h:553-562 as stream events + a deferred WAN tier).
    #
    # push() launches the party-tier collective with async_op=True (RCCL
    # runs it on its comm stream, ordered after the producing compute,
    # so it overlaps the rest of backward — the role of the reference's
    # priority engine pools, threaded_engine_perdevice.cc:82-124) and
    # defers the leader/WAN tier. The first pull (or any state-reading
    # API) flushes every pending key in DESCENDING priority order
    # (ps-lite pops the max meta.priority first, threadsafe_queue.h:50)
    # — deterministic across ranks, so the deferred collectives match.
    # ------------------------------------------------------------------
    def push(self, key, value, priority: int = 0) -> None:
        # list-of-keys form (kvstore.py:162)
        if isinstance(key, (list, tuple)):
            for k, v in zip(key, value):
                self.push(k, v, priority)
            return
        # per-key list of values = this worker's multi-device grads,
        # merged before the wire (the reference groups and sums
        # per-device values, kvstore.py:205 / comm.h Reduce)
        if isinstance(value, (list, tuple)):
            acc = value[0].detach().reshape(-1).float().clone()
            for v in value[1:]:
                acc += v.detach().reshape(-1).float().to(acc.device)
            value = acc
        st = self._state(key)
        if value.numel() != st.numel:
            raise ValueError(f"push size mismatch for {key!r}")
        if key in self._pending:
            # re-push before pull: complete the previous round first
            self.flush()
        st.push_count += 1
        pend = self._party_launch(st, value)
        if self._use_aps():
            # the store-transport tier is ALREADY asynchronous end-to-end
            # (leader hands off and returns); deferring would only delay
            # the hand-off
            self._post_aggregate(key, st, self._party_finish(st, pend))
            return
        self._pending[key] = (priority, self._push_seq, key, pend)
        self._push_seq += 1

    _push_seq = 0

    @property
    def _pending(self):
        p = getattr(self, "_pending_map", None)
        if p is None:
            p = self._pending_map = {}
        return p

    def flush(self) -> None:
        """Complete every pending push: wait the party-tier collectives
        and run the deferred WAN tier + server update, highest priority
        first (P3 ordering)."""
        if not self._pending:
            return
        entries = sorted(self._pending.values(),
                         key=lambda e: (-e[0], e[1]))
        self._pending_map = {}
        for _prio, _seq, key, pend in entries:
            st = self.keys[key]
            party_sum = self._party_finish(st, pend)
            self._post_aggregate(key, st, party_sum)

    def _post_aggregate(self, key, st: _KeyState,
                        party_sum: torch.Tensor) -> None:
        topo = self.topo
        if topo.num_parties == 1:
            if topo.is_leader:
                self._apply_global(key, st, [party_sum])
            return

        if self.cfg.use_hfa:
            self._push_hfa(key, st, party_sum)
            return

        if self._use_aps():
            # true-async: the leader hands its party sum to the global
            # server and returns immediately — no cross-party wait
            if topo.is_leader:
                self._ensure_aps().push(key, party_sum)
            return

        # inter-party (WAN) tier — leaders only
        if topo.is_leader:
            contribs = self._global_exchange_push(key, st, party_sum)
            if contribs is not None:
                self._apply_global(key, st, contribs)

    # -- intra-party tier ------------------------------------------------
    def _party_launch(self, st: _KeyState, value: torch.Tensor):
        """Issue the intra-party aggregation (local server aggregation,
        kvstore_dist_server.h:1286-1296) as an async collective; returns
        a pending record for _party_finish."""
        topo = self.topo
        buf = value.detach().reshape(-1).float().to(self._device)
        if topo.world_size == 1 or topo.num_workers == 1:
            return ("direct", buf.clone(), None, None)
        ctype = self.compression.get("type") if self.compression else None
        if ctype == "2bit":
            # worker-side quantize; leader gathers packed words and
            # dequantize-sums each (CommDevice::ReduceCompressed analog,
            # comm.h:545-590)
            thr = float(self.compression.get("threshold", 0.5))
            if st.residual_2bit is None:
                st.residual_2bit = torch.zeros(st.numel, device=self._device)
            packed = ops.quantize_2bit(buf, st.residual_2bit, thr)
            if topo.backend == "nccl":
                # all_gather is universally supported; party links are xGMI
                out_all = [torch.empty_like(packed)
                           for _ in range(topo.num_workers)]
                work = comm.all_gather(out_all, packed,
                                       group=topo.party_group, async_op=True)
                return ("2bit", out_all, work, thr)
            glist = [torch.empty_like(packed)
                     for _ in range(topo.num_workers)] \
                if topo.is_leader else None
            work = comm.gather(packed, gather_list=glist,
                               dst=topo.leader_rank,
                               group=topo.party_group, async_op=True)
            return ("2bit", glist if glist is not None else buf, work, thr)
        out = buf.clone()
        work = comm.reduce(out, dst=topo.leader_rank, op=dist.ReduceOp.SUM,
                           group=topo.party_group, async_op=True)
        return ("reduce", out, work, None)

    def _party_finish(self, st: _KeyState, pend) -> torch.Tensor:
        kind, payload, work, extra = pend
        if work is not None:
            work.wait()
        if kind == "2bit" and isinstance(payload, list):
            if self.topo.is_leader:
                acc = torch.zeros(st.numel, device=self._device)
                tmp = torch.empty(st.numel, device=self._device)
                for p in payload:
                    ops.dequantize_2bit(p, st.numel, extra, out=tmp)
                    acc += tmp
                return acc
            return payload[0].new_zeros(st.numel, dtype=torch.float32) \
                if payload else torch.zeros(st.numel, device=self._device)
        return payload

    # -- global (WAN) tier ----------------------------------------------
    def _effective_ctype(self, st: _KeyState) -> Optional[str]:
        if not self.compression:
            return None
        ctype = self.compression["type"]
        bound = int(self.compression.get("size_lower_bound",
                                         self.cfg.size_lower_bound))
        if ctype == "mpq":
            # MPQ: small tensors -> fp16, large -> bsc
            # (size gate kvstore_dist_server.h:841,879)
            return "fp16" if st.numel < bound else "bsc"
        if ctype in ("bsc", "bsc_dgt") and st.numel < bound:
            # the reference pushes small tensors PLAIN on the bsc path
            # too (DataPushToGlobalServersBSCompressed size gate, :879);
            # without this a tiny key (a bias) has bsc_capacity 0 and
            # its gradient would vanish entirely
            return None
        return ctype

    def _global_exchange_push(self, key, st: _KeyState,
                              party_sum: torch.Tensor) -> Optional[List[torch.Tensor]]:
        """Leader-tier exchange. Returns the list of per-party
        contributions on ranks that must apply the update (owner in
        sharded mode / all leaders in replicated), else None."""
        topo = self.topo
        P = topo.num_parties
        group = topo.leader_group
        ctype = self._effective_ctype(st)
        ratio = float(self.compression.get("threshold", self.cfg.bsc_ratio)) \
            if self.compression else self.cfg.bsc_ratio

        if ctype in ("bsc", "bsc_dgt"):
            if st.bsc_u is None:
                st.bsc_u = torch.zeros(st.numel, device=self._device)
                st.bsc_v = torch.zeros(st.numel, device=self._device)
            vals, idx = ops.bsc_compress(party_sum, st.bsc_u, st.bsc_v, ratio)
            if ctype == "bsc_dgt":
                # config-5 composition: DGT's 4-bit tier rides the packed
                # VALUES (the reference's DGT chunks whatever bytes a
                # push carries, kv_app.h:917-995); indices stay exact
                dg = self._dgt_state(st, vals.device, numel=vals.numel())
                vals = vals.masked_fill(idx < 0, 0.0)
                dg.residual.zero_()  # packed positions shift every step
                payload = dg.compress(vals) + (idx,)
                gathered = [[torch.empty_like(t) for _ in range(P)]
                            for t in payload]
                for lst, t in zip(gathered, payload):
                    comm.all_gather(lst, t, group=group)
                self.wan.charge(cross_party_bytes(
                    "all_gather", dg.wire_bytes() + idx.numel() * 4, P))
                dense = []
                for pi in range(P):
                    v_p = dg.decompress(*(lst[pi] for lst in gathered[:4]))
                    dense.append(ops.bsc_decompress(
                        v_p.contiguous(), gathered[4][pi], st.numel))
                return dense
            payload = vals.numel() * 4 + idx.numel() * 4
            vlist = [torch.empty_like(vals) for _ in range(P)]
            ilist = [torch.empty_like(idx) for _ in range(P)]
            comm.all_gather(vlist, vals, group=group)
            comm.all_gather(ilist, idx, group=group)
            self.wan.charge(cross_party_bytes("all_gather", payload, P))
            dense = []
            for v_, i_ in zip(vlist, ilist):
                dense.append(ops.bsc_decompress(v_, i_, st.numel))
            return dense

        if ctype == "2bit":
            # the reference also compresses the local->global re-push
            # (DataPushToGlobalServersCompressed,
            # kvstore_dist_server.h:786): leaders keep their own WAN-side
            # residual and exchange packed 2-bit words
            thr = float(self.compression.get("threshold", self.cfg.threshold))
            if st.residual_2bit_wan is None:
                st.residual_2bit_wan = torch.zeros(st.numel,
                                                   device=self._device)
            packed = ops.quantize_2bit(party_sum, st.residual_2bit_wan, thr)
            plist = [torch.empty_like(packed) for _ in range(P)]
            comm.all_gather(plist, packed, group=group)
            self.wan.charge(cross_party_bytes("all_gather",
                                              packed.numel() * 4, P))
            dense = []
            for p_ in plist:
                dense.append(ops.dequantize_2bit(p_, st.numel, thr))
            return dense

        if ctype == "fp16":
            if self._ts is not None:
                # TSEngine relay tree, fp16 wire: 2(P-1) point-to-point
                # hops, no incast link
                return [self._ts.allreduce_sum(party_sum,
                                               wire_dtype=torch.float16)]
            h = party_sum.to(torch.float16)
            hlist = [torch.empty_like(h) for _ in range(P)]
            comm.all_gather(hlist, h, group=group)
            self.wan.charge(cross_party_bytes("all_gather", h.numel() * 2, P))
            return [x.float() for x in hlist]

        if ctype == "dgt":
            dg = self._dgt_state(st, party_sum.device)
            if dg.mode >= 3:
                # real reduced wire: all_gather the 4-bit payload tuple
                # and decompress each party's contribution on arrival
                # (van.cc:750-824 end-to-end, not an emulation charge)
                payload = dg.compress(party_sum)
                gathered = [[torch.empty_like(t) for _ in range(P)]
                            for t in payload]
                for lst, t in zip(gathered, payload):
                    comm.all_gather(lst, t, group=group)
                self.wan.charge(cross_party_bytes(
                    "all_gather", dg.wire_bytes(), P))
                return [dg.decompress(*(lst[p] for lst in gathered))
                        for p in range(P)]
            # modes 1/2: unimportant chunks travel exact (priority-only
            # semantics; no byte saving exists on a reliable fabric)
            contrib_now, wire = dg.transform(party_sum)
            hlist = [torch.empty_like(contrib_now) for _ in range(P)]
            comm.all_gather(hlist, contrib_now, group=group)
            self.wan.charge(cross_party_bytes("all_gather", wire, P))
            return list(hlist)

        if st.sliced and self.cfg.mode == "dist_sync":
            # P3/MultiGPS sliced dense path: each leader owns 1/P of the
            # key; reduce each slice to its owner (a reduce_scatter),
            # update locally, all_gather on pull.
            chunk = st.padded // P
            for c in range(P):
                lo, hi = c * chunk, min(st.numel, (c + 1) * chunk)
                if lo >= hi:
                    continue
                sl = party_sum[lo:hi].clone()
                comm.reduce(sl, dst=topo.leader_ranks[c],
                            op=dist.ReduceOp.SUM, group=group)
                if topo.party_id == c:
                    my_slice = sl
            self.wan.charge(cross_party_bytes("reduce", st.numel * 4, P))
            lo = topo.party_id * chunk
            hi = min(st.numel, lo + chunk)
            if lo < hi:
                w_slice = st.stored[lo:hi]
                if self.optimizer is not None:
                    self.optimizer.update((key, "slice"), w_slice, my_slice)
                else:
                    w_slice.copy_(my_slice)
            return None  # pull all_gathers the slices

        # dense fp32
        if self._ts is not None:
            # TSEngine (replicated dist_sync): scheduler-driven relay
            # merge + spread instead of the all_gather
            return [self._ts.allreduce_sum(party_sum)]
        if self.global_mode == "replicated" or self.cfg.mode == "dist_async":
            flist = [torch.empty_like(party_sum) for _ in range(P)]
            comm.all_gather(flist, party_sum, group=group)
            self.wan.charge(cross_party_bytes("all_gather", st.numel * 4, P))
            return list(flist)
        # sharded dense: reduce to owner (sum), single contribution
        out = party_sum.clone()
        owner_leader = topo.leader_ranks[st.owner_party]
        comm.reduce(out, dst=owner_leader, op=dist.ReduceOp.SUM, group=group)
        self.wan.charge(cross_party_bytes("reduce", st.numel * 4, P))
        if topo.leader_rank == owner_leader and topo.is_leader \
                and topo.party_id == st.owner_party:
            return [out]
        return None

    def _apply_global(self, key, st: _KeyState, contribs: List[torch.Tensor]):
        """Global-server update (ApplyUpdates, kvstore_dist_server.h:535-559).

        dist_sync: one update on the summed contribution.
        dist_async: one sequential update PER party contribution (the
        async server applies each leader push on arrival)."""
        if self.cfg.mode == "dist_async" and self.optimizer is not None \
                and len(contribs) > 1:
            for c in contribs:
                self.optimizer.update(key, st.stored, c)
            st.update_buf = None
            return
        agg = contribs[0]
        for c in contribs[1:]:
            agg = agg + c
        if self.optimizer is not None:
            self.optimizer.update(key, st.stored, agg)
        else:
            # update-on-worker mode: stored holds the aggregated gradient
            st.stored = agg if agg.device == self._device else agg.to(self._device)

    def _push_hfa(self, key, st: _KeyState, party_sum: torch.Tensor):
        """HFA: store party aggregate locally; every K2-th push, exchange
        milestone deltas (stored-milestone)/P and rebase
        (kvstore_dist_server.h:959-972,1324-1343)."""
        topo = self.topo
        if not topo.is_leader:
            return
        st.stored = party_sum
        if st.milestone is None:
            st.milestone = torch.zeros_like(st.stored)
        if st.push_count % self.cfg.hfa_k2 != 0:
            return
        P = topo.num_parties
        delta = (st.stored - st.milestone) / P
        comm.all_reduce(delta, op=dist.ReduceOp.SUM, group=topo.leader_group)
        self.wan.charge(cross_party_bytes("all_reduce", st.numel * 4, P))
        st.stored = st.milestone + delta
        st.milestone = st.stored.clone()

    # ------------------------------------------------------------------
    # pull
    # ------------------------------------------------------------------
    def pull(self, key, out, priority: int = 0,
             ignore_sparse: bool = True) -> None:
        # ignore_sparse accepted for signature parity (kvstore.py:242);
        # row-sparse values live behind row_sparse_pull here.
        # list-of-keys form (kvstore.py:242); a per-key list of outs
        # broadcasts the same value into each (multi-device pull)
        if isinstance(key, (list, tuple)):
            for k, o in zip(key, out):
                self.pull(k, o, priority)
            return
        if isinstance(out, (list, tuple)):
            self.pull(key, out[0], priority)
            for o in out[1:]:
                with torch.no_grad():
                    o.copy_(out[0].to(o.device))
            return
        self.flush()
        st = self._state(key)
        topo = self.topo
        if self._use_aps():
            if topo.is_leader:
                st.stored = self._ensure_aps().pull(key)
        else:
            self._global_exchange_pull(key, st)
        # intra-party: leader broadcasts authoritative value to its workers
        if topo.world_size > 1 and topo.num_workers > 1:
            comm.broadcast(st.stored, src=topo.leader_rank,
                           group=topo.party_group)
        result = st.stored.reshape(st.shape).to(out.dtype)
        with torch.no_grad():
            out.reshape(st.shape).copy_(result)

    def _global_exchange_pull(self, key, st: _KeyState):
        """Owner -> leaders distribution of the updated value (pull from
        global servers, kvstore_dist_server.h:899-1094). No-op in
        replicated mode / single party / HFA (already rebased)."""
        topo = self.topo
        P = topo.num_parties
        if P == 1 or self.cfg.use_hfa:
            return
        ctype = self._effective_ctype(st)
        need_wire = self.global_mode == "sharded" and not (
            ctype in ("bsc", "fp16", "dgt", "2bit", "bsc_dgt")
            or self.cfg.mode == "dist_async")
        if not need_wire:
            return  # all leaders already hold the result (replayed update)
        if not topo.is_leader:
            return
        group = topo.leader_group
        owner_leader = topo.leader_ranks[st.owner_party]
        if st.sliced:
            # all_gather the per-leader slices (MultiGPS reassembly,
            # kvstore_dist_server.h:1039-1094 — here order is by
            # construction, no sort needed)
            chunk = st.padded // P
            lo = topo.party_id * chunk
            hi = min(st.numel, lo + chunk)
            mine = torch.zeros(chunk, device=self._device)
            if lo < hi:
                mine[:hi - lo] = st.stored[lo:hi]
            parts = [torch.empty_like(mine) for _ in range(P)]
            comm.all_gather(parts, mine, group=group)
            self.wan.charge(cross_party_bytes("all_gather", chunk * 4, P))
            st.stored = torch.cat(parts)[:st.numel]
            return
        # NOTE on pull-side BSC (BSCPullCompress,
        # gradient_compression.cc:271-308, response sizing
        # kvstore_dist_server.h:1194): the reference needs it because
        # its global server is a star — the aggregated gradient must
        # travel BACK over the WAN to every local server. Here every
        # compressed exchange is an all_gather of the per-party
        # payloads, so all leaders already hold the aggregate and the
        # pull direction has NO wire at all (need_wire above) — the
        # "bi-directional" saving is structural. The op itself
        # (ops.bsc_pull_compress + k_bsc_pull_pack) remains for star
        # topologies and is golden-tested against the reference
        # semantics on CPU and gfx950.
        comm.broadcast(st.stored, src=owner_leader, group=group)
        self.wan.charge(cross_party_bytes("broadcast", st.numel * 4, P))

    def push_row_sparse(self, key, row_ids: torch.Tensor,
                        values: torch.Tensor, priority: int = 0) -> None:
        """Push a row-sparse gradient (rows named by row_ids, one value
        row each — the reference's row_sparse push storage,
        kvstore_dist.h:900 EncodeRowSparseKey). Duplicate ids
        accumulate. Densified before the wire: in-node xGMI bandwidth
        makes dense collectives the faster transport for these sizes."""
        st = self._state(key)
        if len(st.shape) < 2:
            raise ValueError("push_row_sparse needs a >=2d key")
        rows = st.shape[0]
        width = st.numel // rows
        dense = torch.zeros(rows, width, dtype=torch.float32,
                            device=self._device)
        dense.index_add_(0, row_ids.to(self._device).long(),
                         values.detach().reshape(-1, width).float()
                         .to(self._device))
        self.push(key, dense, priority)

    def _comm_device(self) -> torch.device:
        return torch.device("cpu") if self.topo.backend == "gloo" \
            else self._device

    def row_sparse_pull(self, key, out, row_ids,
                        priority: int = 0) -> None:
        """Pull only the rows named by row_ids (python/mxnet/kvstore.py:316).

        Rows-only wire (EncodeRowSparseKey kvstore_dist.h:900 semantics):
        each worker ships its row-id list to its party leader and
        receives exactly those rows back — worker-tier bytes are
        proportional to len(row_ids), not to the key size. On the leader
        (WAN) tier, sharded-dense keys fetch only the union of the
        party's requested rows from the owner; replicated / compressed /
        HFA modes hold the value on every leader already and need no
        global wire at all. Sliced (P3/MultiGPS) keys reassemble on the
        leader first (rows stripe across every leader's chunk), the
        worker tier still moves rows only. Row-id dedup is torch.unique
        (the reference used cub Unique, kvstore_utils.cu:44-111).

        `out` must be [len(row_ids), row_width]; `out` and `row_ids`
        may be aligned lists (multi-device form)."""
        if isinstance(out, (list, tuple)):
            if not isinstance(row_ids, (list, tuple)):
                row_ids = [row_ids] * len(out)
            for o, r in zip(out, row_ids):
                self.row_sparse_pull(key, o, r, priority)
            return
        self.flush()
        st = self._state(key)
        if len(st.shape) < 2:
            raise ValueError("row_sparse_pull needs a >=2d key")
        rows = st.shape[0]
        width = st.numel // rows
        topo = self.topo
        ids = row_ids.reshape(-1).long()

        if topo.world_size == 1:
            sel = st.stored.reshape(rows, width)[ids.to(self._device)]
            with torch.no_grad():
                out.reshape(ids.numel(), width).copy_(sel.to(out.dtype))
            return

        cdev = self._comm_device()
        # --- party tier, phase 1: workers -> leader row-id lists -------
        my_ids = torch.unique(ids).to(cdev)
        if topo.num_workers > 1:
            counts = [torch.zeros(1, dtype=torch.int64, device=cdev)
                      for _ in range(topo.num_workers)]
            cnt = torch.tensor([my_ids.numel()], dtype=torch.int64,
                               device=cdev)
            comm.all_gather(counts, cnt, group=topo.party_group)
            if topo.is_leader:
                peer_ids = {}
                for i, r in enumerate(topo.party_ranks):
                    if r == topo.rank:
                        continue
                    buf = torch.empty(int(counts[i].item()),
                                      dtype=torch.int64, device=cdev)
                    comm.recv(buf, src=r)
                    peer_ids[r] = buf
            else:
                comm.send(my_ids, dst=topo.leader_rank)

        # --- leader (WAN) tier: make the requested rows authoritative --
        if topo.is_leader:
            if topo.num_workers > 1:
                union = torch.unique(torch.cat(
                    [my_ids] + list(peer_ids.values())))
            else:
                union = my_ids
            self._global_pull_rows(key, st, union.to(self._device),
                                   rows, width)

        # --- party tier, phase 2: leader -> workers selected rows ------
        if topo.num_workers > 1:
            if topo.is_leader:
                stored2d = st.stored.reshape(rows, width)
                for i, r in enumerate(topo.party_ranks):
                    if r == topo.rank:
                        continue
                    sel = stored2d[peer_ids[r].to(self._device)] \
                        .to(cdev).contiguous()
                    comm.send(sel, dst=r)
                sel = stored2d[ids.to(self._device)]
            else:
                rbuf = torch.empty(my_ids.numel(), width, device=cdev)
                comm.recv(rbuf, src=topo.leader_rank)
                # scatter unique rows back to the (possibly repeated)
                # requested order
                pos = torch.searchsorted(my_ids.cpu(), ids.cpu())
                sel = rbuf[pos.to(cdev)]
        else:
            sel = st.stored.reshape(rows, width)[ids.to(self._device)]
        with torch.no_grad():
            out.reshape(ids.numel(), width).copy_(sel.to(out.device,
                                                         out.dtype))

    def _global_pull_rows(self, key, st: _KeyState, union: torch.Tensor,
                          rows: int, width: int) -> None:
        """Leader-tier row fetch. Only the sharded dense mode has a wire
        here (the owner holds the authoritative value); everything else
        already replays the update on every leader."""
        topo = self.topo
        P = topo.num_parties
        if P == 1 or self.cfg.use_hfa:
            return
        if self._use_aps():
            if topo.is_leader:
                st.stored = self._ensure_aps().pull(key)
            return
        ctype = self._effective_ctype(st)
        need_wire = self.global_mode == "sharded" and not (
            ctype in ("bsc", "fp16", "dgt", "2bit", "bsc_dgt")
            or self.cfg.mode == "dist_async")
        if not need_wire:
            return
        if st.sliced:
            # rows stripe across every leader's chunk: reassemble once
            self._global_exchange_pull(key, st)
            return
        cdev = self._comm_device()
        owner_leader = topo.leader_ranks[st.owner_party]
        if topo.rank == owner_leader:
            stored2d = st.stored.reshape(rows, width)
            for r in topo.leader_ranks:
                if r == owner_leader:
                    continue
                cnt = torch.zeros(1, dtype=torch.int64, device=cdev)
                comm.recv(cnt, src=r)
                idx = torch.empty(int(cnt.item()), dtype=torch.int64,
                                  device=cdev)
                comm.recv(idx, src=r)
                sel = stored2d[idx.to(self._device)].to(cdev).contiguous()
                comm.send(sel, dst=r)
                self.wan.charge(sel.numel() * 4 + idx.numel() * 8)
        else:
            u = union.to(cdev)
            comm.send(torch.tensor([u.numel()], dtype=torch.int64,
                                   device=cdev), dst=owner_leader)
            comm.send(u, dst=owner_leader)
            rbuf = torch.empty(u.numel(), width, device=cdev)
            comm.recv(rbuf, src=owner_leader)
            st.stored.reshape(rows, width)[union.to(self._device)] = \
                rbuf.to(self._device)

    # ------------------------------------------------------------------
    # checkpointing (layout parity: named-param dict + separate optimizer
    # state blob — gluon/block.py:315,356 + kvstore.py:566-592)
    # ------------------------------------------------------------------
    def save_optimizer_states(self, fname: str, dump_optimizer: bool = False):
        """Serialize optimizer STATES; with dump_optimizer=True also the
        optimizer itself (the spec), matching the reference where the
        pickled optimizer rides along only on request
        (python/mxnet/kvstore.py:566-592)."""
        self.flush()
        if self.optimizer is None:
            raise RuntimeError("no optimizer set")
        blob = self.optimizer.state_dict()
        if not dump_optimizer:
            blob = {k: v for k, v in blob.items() if k != "spec"}
        with open(fname, "wb") as f:
            pickle.dump(blob, f)

    def load_optimizer_states(self, fname: str):
        with open(fname, "rb") as f:
            blob = pickle.load(f)
        if self.optimizer is None:
            if "spec" not in blob:
                raise RuntimeError(
                    "states-only blob (saved with dump_optimizer=False) "
                    "needs set_optimizer() before load_optimizer_states")
            self.optimizer = ServerOptimizer(OptimizerSpec(**blob["spec"]))
        self.optimizer.load_state_dict(blob, device=self._device)

    # ------------------------------------------------------------------
    # DGT transform (shared DGTState, kvstore/dgt.py)
    # ------------------------------------------------------------------
    def _dgt_state(self, st: _KeyState, device, numel=None):
        if not hasattr(self, "_dgt_states"):
            self._dgt_states = {}
        n = numel if numel is not None else st.numel
        key = (id(st), n)
        dg = self._dgt_states.get(key)
        if dg is None or dg.numel != n:
            from .dgt import DGTState
            dg = DGTState(n, device,
                          chunk_elems=max(64, self.cfg.dgt_block_size // 4),
                          k=self.cfg.dgt_k, alpha=self.cfg.dgt_alpha,
                          mode=self.cfg.enable_dgt or 3)
            self._dgt_states[key] = dg
        return dg
