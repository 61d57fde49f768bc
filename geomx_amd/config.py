"""Configuration for the geomx_amd hierarchical parameter-server framework.

Single Config dataclass fronting environment variables. Keeps GeoMX's
DMLC_* env-var compatibility for topology where sensible (the reference
parses roles/topology from env in ps-lite: postoffice.cc:22-53 and
docs/source/env-var-summary.rst), and GeoMX feature toggles
(ENABLE_DGT, MXNET_KVSTORE_USE_HFA, MXNET_KVSTORE_HFA_K1/K2,
MXNET_KVSTORE_SIZE_LOWER_BOUND — kv_app.h:842-850,
kvstore_dist_server.h:182-187) under their original names plus GEOMX_*
aliases.

MI355X-native topology: there are no separate scheduler/server
processes. Every rank is a worker; rank 0 of each party carries the
"local server" role; the global-server state is sharded across party
leaders (MultiGPS, kvstore_dist_server.h:1770-1810).
"""

from __future__ import annotations

import dataclasses
import os
from typing import List, Optional


def _env_int(names, default):
    for n in names if isinstance(names, (list, tuple)) else [names]:
        v = os.environ.get(n)
        if v is not None and v != "":
            return int(v)
    return default


def _env_float(names, default):
    for n in names if isinstance(names, (list, tuple)) else [names]:
        v = os.environ.get(n)
        if v is not None and v != "":
            return float(v)
    return default


def _env_str(names, default):
    for n in names if isinstance(names, (list, tuple)) else [names]:
        v = os.environ.get(n)
        if v is not None and v != "":
            return v
    return default


@dataclasses.dataclass
class Config:
    # --- topology -----------------------------------------------------
    # World size / rank come from torch.distributed env (RANK/WORLD_SIZE).
    # num_parties partitions the world into contiguous "data centers";
    # party i has ranks [i*party_size, (i+1)*party_size).
    num_parties: int = 1
    # explicit party sizes (sum == world_size); empty -> uniform split
    party_sizes: Optional[List[int]] = None

    # --- synchronization algorithm ------------------------------------
    # "dist_sync"  -> FSA: both tiers synchronous (kvstore.cc:55-62)
    # "dist_async" -> MixedSync: async global tier (DataHandleAsyncDefault)
    mode: str = "dist_sync"
    # dist_async transport: "lockstep" reproduces the async UPDATE MATH
    # over synchronous collectives; "store" is the true-async
    # parameter server (kvstore/async_ps.py) where parties proceed at
    # their own pace
    async_transport: str = "lockstep"
    # HFA (hierarchical frequency aggregation, examples/cnn_hfa.py):
    # workers run K1 local steps between pushes; leaders forward to the
    # global tier only every K2-th aggregation (kvstore_dist_server.h:1324-1343)
    use_hfa: bool = False
    hfa_k1: int = 1
    hfa_k2: int = 1

    # --- compression --------------------------------------------------
    # None | "2bit" | "bsc" | "fp16" | "mpq"
    compression: Optional[str] = None
    # 2bit threshold (gradient_compression-inl.h:40-95)
    threshold: float = 0.5
    # bsc compression ratio (gradient_compression.cc:191-269)
    bsc_ratio: float = 0.01
    # MPQ size gate: tensors with numel < size_lower_bound take the fp16
    # path, larger ones the bsc path (kvstore_dist_server.h:841,879)
    size_lower_bound: int = 200_000

    # --- DGT (differential gradient transmission) ---------------------
    # 0=off, 1..3 map the reference's channel modes (van.cc:736-748) to
    # priority-ordered buckets on side streams: important buckets go
    # first at full precision, unimportant buckets late (and 4-bit
    # quantized at enable_dgt==3).
    enable_dgt: int = 0
    dgt_k: float = 0.5           # fraction of chunks deemed important (DMLC_K)
    dgt_k_min: float = 0.2       # DMLC_K_MIN (lower bound; ADAPTIVE_K_FLAG is
                                 # parsed but never acted on in the reference —
                                 # kv_app.h:848 dead config — kept for parity)
    dgt_channels: int = 1        # DMLC_UDP_CHANNEL_NUM (priority tiers)
    dgt_block_size: int = 4096   # bytes per chunk (DGT_BLOCK_SIZE)
    dgt_alpha: float = 0.3       # EWMA contribution factor (DGT_CONTRIBUTION_ALPHA)

    # TSEngine toggle (ENABLE_INTER_TS / ENABLE_INTRA_TS): selects the
    # incast-free replicated global tier (see kvstore/dist.py docstring)
    enable_ts: bool = False
    # P3 toggle (ENABLE_P3): priority scheduling is structural here
    # (reverse-order buckets; big-key slicing via bigarray_bound), the
    # flag is accepted for launch-script compatibility
    enable_p3: bool = False

    # --- WAN emulation -------------------------------------------------
    # Bandwidth cap (Gbit/s) applied to inter-party (leader<->leader)
    # traffic via a token bucket, so Bi-Sparse/MPQ/DGT speedups are
    # measurable in-node. 0 = uncapped.
    wan_gbps: float = 0.0
    # Optional per-party uplink rates (Gbit/s, one per party) for
    # HETEROGENEOUS WAN emulation — the regime TSEngine's relay
    # scheduling targets (a slow data center's link is visited exactly
    # once per relay instead of sitting on the collective's ring).
    # Overrides wan_gbps for each listed party; GEOMX_PARTY_WAN_GBPS is
    # the comma-separated env form.
    party_wan_gbps: Optional[List[float]] = None
    # Propagation delay (round-trip, ms) added per emulated transfer —
    # inter-DC WANs are tens of ms; deep relay overlays pay it per hop.
    wan_rtt_ms: float = 0.0

    # --- runtime -------------------------------------------------------
    # P3/MultiGPS big-tensor slicing: keys with numel >= bigarray_bound
    # are sliced across party leaders (EncodeP3Key / bigarray_bound_,
    # kvstore_dist.h:69,763-799; MXNET_KVSTORE_BIGARRAY_BOUND)
    bigarray_bound: int = 1_000_000
    bucket_mb: int = 25          # gradient bucket size for fused collectives
    # wire dtype for the flat-mode bucket all_reduce: "fp32" (exact) or
    # "bf16" (half traffic — the FP16-transmission feature applied to the
    # fast path; server/optimizer math stays fp32)
    comm_dtype: str = "fp32"
    overlap: bool = True         # overlap comm with backward
    backend: Optional[str] = None  # override; default nccl on GPU, gloo on CPU
    device: Optional[str] = None

    @staticmethod
    def from_env(**overrides) -> "Config":
        cfg = Config(
            num_parties=_env_int(["GEOMX_NUM_PARTIES"], 1),
            mode=_env_str(["GEOMX_MODE"], "dist_sync"),
            use_hfa=bool(_env_int(["MXNET_KVSTORE_USE_HFA", "GEOMX_USE_HFA"], 0)),
            hfa_k1=_env_int(["MXNET_KVSTORE_HFA_K1", "GEOMX_HFA_K1"], 1),
            hfa_k2=_env_int(["MXNET_KVSTORE_HFA_K2", "GEOMX_HFA_K2"], 1),
            compression=_env_str(["GEOMX_COMPRESSION"], None),
            bsc_ratio=_env_float(["GEOMX_BSC_RATIO"], 0.01),
            threshold=_env_float(["GEOMX_2BIT_THRESHOLD"], 0.5),
            size_lower_bound=_env_int(
                ["MXNET_KVSTORE_SIZE_LOWER_BOUND", "GEOMX_SIZE_LOWER_BOUND"], 200_000
            ),
            enable_dgt=_env_int(["ENABLE_DGT", "GEOMX_ENABLE_DGT"], 0),
            dgt_k=_env_float(["DMLC_K", "GEOMX_DGT_K"], 0.5),
            dgt_k_min=_env_float(["DMLC_K_MIN"], 0.2),
            dgt_channels=_env_int(["DMLC_UDP_CHANNEL_NUM"], 1),
            enable_ts=bool(_env_int(["ENABLE_INTER_TS"], 0)
                           or _env_int(["ENABLE_INTRA_TS"], 0)),
            enable_p3=bool(_env_int(["ENABLE_P3"], 0)),
            dgt_block_size=_env_int(["DGT_BLOCK_SIZE", "GEOMX_DGT_BLOCK_SIZE"], 4096),
            dgt_alpha=_env_float(["DGT_CONTRIBUTION_ALPHA", "GEOMX_DGT_ALPHA"], 0.3),
            wan_gbps=_env_float(["GEOMX_WAN_GBPS"], 0.0),
            wan_rtt_ms=_env_float(["GEOMX_WAN_RTT_MS"], 0.0),
            bucket_mb=_env_int(["GEOMX_BUCKET_MB"], 25),
            bigarray_bound=_env_int(
                ["MXNET_KVSTORE_BIGARRAY_BOUND", "GEOMX_BIGARRAY_BOUND"],
                1_000_000),
        )
        sizes = _env_str(["GEOMX_PARTY_SIZES"], None)
        if sizes:
            cfg.party_sizes = [int(x) for x in sizes.split(",") if x]
        rates = _env_str(["GEOMX_PARTY_WAN_GBPS"], None)
        if rates:
            cfg.party_wan_gbps = [float(x) for x in rates.split(",") if x]
        for k, v in overrides.items():
            if not hasattr(cfg, k):
                raise ValueError(f"unknown config field {k!r}")
            setattr(cfg, k, v)
        cfg.validate()
        return cfg

    def validate(self):
        if self.mode not in ("dist_sync", "dist_async", "local"):
            raise ValueError(f"mode must be dist_sync|dist_async|local, got {self.mode}")
        if self.compression not in (None, "2bit", "bsc", "fp16", "mpq", "dgt", "bsc_dgt"):
            raise ValueError(f"unknown compression {self.compression!r}")
        if not (0 < self.bsc_ratio < 1):
            raise ValueError("bsc_ratio must be in (0,1)")
        if self.num_parties < 1:
            raise ValueError("num_parties >= 1")
        if self.hfa_k1 < 1 or self.hfa_k2 < 1:
            raise ValueError("hfa_k1/k2 >= 1")
        if self.async_transport not in ("lockstep", "store"):
            raise ValueError("async_transport must be lockstep|store")
        if self.party_wan_gbps is not None \
                and len(self.party_wan_gbps) != self.num_parties:
            raise ValueError("party_wan_gbps needs one rate per party")
        return self

    def wan_rate_for(self, party_id: int) -> float:
        """This party's emulated WAN uplink rate (Gbit/s; 0 = uncapped).
        Heterogeneous per-party rates override the global wan_gbps."""
        if self.party_wan_gbps is not None:
            return float(self.party_wan_gbps[party_id])
        return self.wan_gbps
