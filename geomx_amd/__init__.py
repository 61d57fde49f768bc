"""geomx_amd — an MI355X-native hierarchical parameter-server training
framework with the capabilities of INET-RC/GeoMX.

Brand-new design for AMD Instinct MI355X (gfx950 / CDNA4):
  - PyTorch-ROCm frontend, one process per GPU, RCCL over xGMI
  - HiPS two-level aggregation as nested process groups (party tier +
    leader/WAN tier) instead of ps-lite message passing
  - hand-written HIP kernels for the kvstore hot path (2bit, Bi-Sparse,
    FP16/MPQ, DGT, fused SGD/Adam/DCASGD) in the `_geops` extension
  - WAN emulation via a token-bucket bandwidth cap on inter-party links

Reference (behavioral parity only): https://github.com/INET-RC/GeoMX
"""

__version__ = "0.1.0"

from . import config, topology  # noqa: F401
from .config import Config  # noqa: F401
from .kvstore import create  # noqa: F401
from .kvstore.optimizer import OptimizerSpec  # noqa: F401

# mx.kv.create parity alias
class kv:  # noqa: N801
    create = staticmethod(create)
