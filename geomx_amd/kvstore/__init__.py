"""geomx_amd.kvstore — GeoMX-compatible kvstore factory.

`create("dist_sync")` / `create("dist_async")` mirrors
python/mxnet/kvstore.py:663 + src/kvstore/kvstore.cc:41.
"""

from __future__ import annotations

from typing import Optional

from ..config import Config
from ..topology import Topology
from .base import KVStoreBase
from .dist import KVStoreDist
from .optimizer import OptimizerSpec, ServerOptimizer
from .wan import TokenBucket, cross_party_bytes

__all__ = ["create", "KVStoreBase", "KVStoreDist", "OptimizerSpec",
           "ServerOptimizer", "TokenBucket", "cross_party_bytes"]


def create(name: str = "dist_sync", cfg: Optional[Config] = None,
           topo: Optional[Topology] = None, **overrides) -> KVStoreDist:
    """Create a kvstore.

    name: "dist_sync" (FSA), "dist_async" (MixedSync), or "local".
    The reference resolves types by SUBSTRING (kvstore.cc:41-80:
    "dist" selects the distributed store, "_sync" within it selects
    synchronous mode, "device"/"nccl" pick intra-node comm variants) —
    so names like "dist_device_sync", "device", "nccl" are accepted
    here the same way. Device/NCCL distinctions collapse on MI355X:
    RCCL over xGMI IS the (only) intra-node fabric.

    Extra keyword arguments override Config fields (which themselves
    default from the GeoMX-compatible environment variables).
    """
    global_mode = overrides.pop("global_mode", None)
    if cfg is None:
        cfg = Config.from_env(**overrides)
    if global_mode is None:
        # ENABLE_INTER_TS/ENABLE_INTRA_TS select the incast-free
        # replicated tier (the TSEngine role)
        global_mode = "replicated" if cfg.enable_ts else "sharded"
    t = str(name).lower()
    if "dist" in t:
        cfg.mode = "dist_sync" if "_sync" in t else "dist_async"
    else:  # local / device / nccl: single-node synchronous store
        cfg.mode = "dist_sync"
    cfg.validate()
    kv = KVStoreDist(cfg, topo=topo, global_mode=global_mode)
    kv._type_name = t
    return kv
