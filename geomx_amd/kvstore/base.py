"""Abstract kvstore interface mirroring GeoMX's Python API surface.

Reference: /root/reference/python/mxnet/kvstore.py:99-661 (class KVStore:
init/push/pull/set_optimizer/set_gradient_compression/rank/num_workers/
num_all_workers/is_master_worker/save_optimizer_states/
load_optimizer_states/_barrier) and include/mxnet/kvstore.h:59-352.
"""

from __future__ import annotations

from typing import Dict

import torch


class KVStoreBase:
    def init(self, key, value: torch.Tensor) -> None:
        raise NotImplementedError

    def push(self, key, value: torch.Tensor, priority: int = 0) -> None:
        raise NotImplementedError

    def pull(self, key, out: torch.Tensor, priority: int = 0) -> None:
        raise NotImplementedError

    def set_optimizer(self, optimizer) -> None:
        raise NotImplementedError

    def set_gradient_compression(self, compression_params: Dict) -> None:
        raise NotImplementedError

    def barrier(self) -> None:
        raise NotImplementedError

    # GeoMX exposes _barrier; keep the alias
    _barrier = barrier

    def wait_all(self) -> None:
        """Engine-drain analog of mx.nd.waitall()."""
        if torch.cuda.is_available():
            torch.cuda.synchronize()

    @property
    def type(self) -> str:
        raise NotImplementedError

    @property
    def rank(self) -> int:
        raise NotImplementedError

    @property
    def num_workers(self) -> int:
        raise NotImplementedError

    @property
    def num_all_workers(self) -> int:
        raise NotImplementedError

    @property
    def is_master_worker(self) -> bool:
        raise NotImplementedError

    def save_optimizer_states(self, fname: str, dump_optimizer: bool = False) -> None:
        raise NotImplementedError

    def load_optimizer_states(self, fname: str) -> None:
        raise NotImplementedError
