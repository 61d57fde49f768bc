"""Server-side optimizer: the ApplyUpdates analog.

GeoMX pickles a Python optimizer to the global server, which runs it on
the aggregated gradient (kvstore_dist_server.h:535-559; updater runner
Executor :109-168; set_optimizer python/mxnet/kvstore.py:452-500).

Here the "global server" is the party-leader group; each leader owns a
key shard and applies a FUSED HIP update kernel (geomx_amd/ops) to its
authoritative fp32 copy. OptimizerSpec is the picklable equivalent of
the reference's pickled optimizer object.
"""

from __future__ import annotations

import dataclasses
from typing import Dict, Optional

import torch

from .. import ops


@dataclasses.dataclass
class OptimizerSpec:
    name: str = "sgd"   # sgd|sgd_mom|adam|dcasgd|rmsprop|adagrad|signsgd|signum
    lr: float = 0.01
    momentum: float = 0.9
    beta1: float = 0.9
    beta2: float = 0.999
    eps: float = 1e-8
    wd: float = 0.0
    lamda: float = 0.04          # DCASGD delay-compensation scale
    rho: float = 0.9             # RMSProp decay
    rescale_grad: float = 1.0
    clip_gradient: Optional[float] = None  # clamp |g| per element AFTER
                                           # rescale (optimizer.py:912 —
                                           # every reference optimizer
                                           # clips this way)

    def validate(self):
        if self.name not in ("sgd", "sgd_mom", "adam", "dcasgd", "rmsprop",
                             "adagrad", "signsgd", "signum"):
            raise ValueError(f"unknown optimizer {self.name!r}")
        return self


class ServerOptimizer:
    """Per-key optimizer state + fused update dispatch on the owner rank."""

    @property
    def learning_rate(self) -> float:
        return self.spec.lr

    def set_learning_rate(self, lr: float) -> None:
        """Runtime LR update (gluon Trainer.set_learning_rate /
        Optimizer lr_scheduler parity): takes effect on the next
        update; optimizer state is untouched."""
        self.spec.lr = float(lr)

    def __init__(self, spec: OptimizerSpec):
        self.spec = spec.validate()
        self.state: Dict[object, Dict[str, torch.Tensor]] = {}
        self.step_count: Dict[object, int] = {}

    def _get_state(self, key, w: torch.Tensor) -> Dict[str, torch.Tensor]:
        st = self.state.get(key)
        if st is None:
            st = {}
            if self.spec.name == "sgd_mom":
                st["mom"] = torch.zeros_like(w)
            elif self.spec.name == "adam":
                st["m"] = torch.zeros_like(w)
                st["v"] = torch.zeros_like(w)
            elif self.spec.name == "dcasgd":
                st["prev_w"] = w.clone()
                if self.spec.momentum != 0.0:
                    st["mom"] = torch.zeros_like(w)
            elif self.spec.name in ("rmsprop", "adagrad"):
                st["n"] = torch.zeros_like(w)
            elif self.spec.name == "signum":
                st["mom"] = torch.zeros_like(w)
            self.state[key] = st
            self.step_count[key] = 0
        return st

    def update(self, key, w: torch.Tensor, grad: torch.Tensor,
               rescale: Optional[float] = None):
        """Apply one fused update of `w` (fp32, flat) with aggregated grad."""
        s = self.spec
        rs = s.rescale_grad if rescale is None else rescale
        if s.clip_gradient is not None:
            # reference clips the RESCALED gradient, so fold the scale
            # in here and hand the kernels rs=1
            grad = torch.clamp(grad * rs, -s.clip_gradient,
                               s.clip_gradient)
            rs = 1.0
        st = self._get_state(key, w)
        self.step_count[key] += 1
        t = self.step_count[key]
        if s.name == "sgd":
            ops.sgd_update(w, grad, s.lr, s.wd, rs)
        elif s.name == "sgd_mom":
            ops.sgd_mom_update(w, grad, st["mom"], s.lr, s.momentum, s.wd, rs)
        elif s.name == "adam":
            ops.adam_update(w, grad, st["m"], st["v"], t, s.lr, s.beta1,
                            s.beta2, s.eps, s.wd, rs)
        elif s.name == "dcasgd":
            ops.dcasgd_update(w, grad, st["prev_w"], st.get("mom"), s.lr,
                              s.lamda, s.momentum, s.wd, rs)
        elif s.name == "rmsprop":
            ops.rmsprop_update(w, grad, st["n"], s.lr, s.rho, s.eps, s.wd, rs)
        elif s.name == "adagrad":
            ops.adagrad_update(w, grad, st["n"], s.lr, s.eps, s.wd, rs)
        elif s.name == "signsgd":
            ops.signsgd_update(w, grad, s.lr, s.wd, rs)
        elif s.name == "signum":
            ops.signum_update(w, grad, st["mom"], s.lr, s.momentum, s.wd, rs)

    # -- checkpointing (kvstore.save_optimizer_states layout: a separate
    #    optimizer-state blob, python/mxnet/kvstore.py:566-592) ----------
    def state_dict(self) -> Dict:
        return {
            "spec": dataclasses.asdict(self.spec),
            "step_count": dict(self.step_count),
            "state": {k: {n: t.cpu() for n, t in st.items()}
                      for k, st in self.state.items()},
        }

    def load_state_dict(self, d: Dict, device=None):
        if "spec" in d:  # absent when saved with dump_optimizer=False
            self.spec = OptimizerSpec(**d["spec"]).validate()
        self.step_count = dict(d["step_count"])
        self.state = {}
        for k, st in d["state"].items():
            self.state[k] = {n: (t.to(device) if device else t.clone())
                             for n, t in st.items()}
