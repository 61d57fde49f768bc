"""Checkpoint / resume with the GeoMX layout: a named-parameter dict
file plus a SEPARATE optimizer-state blob.

Parity targets:
  - Block.save_parameters/load_parameters (gluon/block.py:315,356):
    dict of named arrays in one file.
  - kvstore.save_optimizer_states/load_optimizer_states
    (python/mxnet/kvstore.py:566-592): optimizer state saved separately
    (in HiPS the global server owns it; here the leader/ServerOptimizer).
"""

from __future__ import annotations

from typing import Dict, Optional

import torch


def save_parameters(model: torch.nn.Module, fname: str) -> None:
    """Named-param dict, values on CPU (the reference serializes NDArray
    dicts; we use torch.save of a plain {name: tensor} dict)."""
    sd = {k: v.detach().cpu() for k, v in model.state_dict().items()}
    torch.save(sd, fname)


def load_parameters(model: torch.nn.Module, fname: str,
                    device: Optional[torch.device] = None,
                    strict: bool = True) -> None:
    sd = torch.load(fname, map_location="cpu", weights_only=True)
    model.load_state_dict(sd, strict=strict)
    if device is not None:
        model.to(device)


def save_checkpoint(model: torch.nn.Module, kvstore_or_trainer, prefix: str,
                    epoch: int) -> Dict[str, str]:
    """Module.save_checkpoint analog (module/module.py:165):
    `prefix-NNNN.params` + `prefix-NNNN.states`."""
    pfile = f"{prefix}-{epoch:04d}.params"
    sfile = f"{prefix}-{epoch:04d}.states"
    save_parameters(model, pfile)
    saver = getattr(kvstore_or_trainer, "save_optimizer_states", None)
    if saver is not None:
        saver(sfile, dump_optimizer=True)
    else:  # GeoTrainer: its ServerOptimizer holds the state
        import pickle
        with open(sfile, "wb") as f:
            pickle.dump(kvstore_or_trainer.server_opt.state_dict(), f)
    return {"params": pfile, "states": sfile}


def load_checkpoint(model: torch.nn.Module, kvstore_or_trainer, prefix: str,
                    epoch: int, device: Optional[torch.device] = None) -> None:
    pfile = f"{prefix}-{epoch:04d}.params"
    sfile = f"{prefix}-{epoch:04d}.states"
    load_parameters(model, pfile, device=device)
    loader = getattr(kvstore_or_trainer, "load_optimizer_states", None)
    if loader is not None:
        loader(sfile)
    else:
        import pickle
        with open(sfile, "rb") as f:
            kvstore_or_trainer.server_opt.load_state_dict(
                pickle.load(f), device=device)
    # trainer keeps params flattened: refresh the flat buffers
    refresh = getattr(kvstore_or_trainer, "refresh_params", None)
    if refresh is not None:
        refresh()
