"""Per-iteration measurement logging (examples/utils.py Measure analog:
wall-clock + per-phase iteration timings dumped as JSON) and simple
training metrics (python/mxnet/metric.py Accuracy/CrossEntropy analog).
"""

from __future__ import annotations

import json
import time
from typing import Dict, List, Optional

import torch


class Measure:
    """Phase timer: measure.start('fwd') ... measure.stop('fwd') per
    iteration; dump() writes per-iteration JSON rows."""

    def __init__(self, sync_cuda: bool = True):
        self.sync_cuda = sync_cuda and torch.cuda.is_available()
        self.rows: List[Dict] = []
        self._cur: Dict[str, float] = {}
        self._open: Dict[str, float] = {}
        self._t0 = time.time()

    def _now(self):
        if self.sync_cuda:
            torch.cuda.synchronize()
        return time.perf_counter()

    def start(self, phase: str):
        self._open[phase] = self._now()

    def stop(self, phase: str):
        t = self._now()
        self._cur[phase] = self._cur.get(phase, 0.0) + (t - self._open.pop(phase))

    def next_iteration(self, **extra):
        row = {"wall": time.time() - self._t0, **self._cur, **extra}
        self.rows.append(row)
        self._cur = {}
        return row

    def dump(self, fname: str):
        with open(fname, "w") as f:
            for row in self.rows:
                f.write(json.dumps(row) + "\n")


class Accuracy:
    def __init__(self):
        self.correct = 0
        self.total = 0

    def update(self, labels: torch.Tensor, preds: torch.Tensor):
        if preds.dim() > 1:
            preds = preds.argmax(dim=-1)
        self.correct += (preds == labels).sum().item()
        self.total += labels.numel()

    def get(self) -> float:
        return self.correct / max(1, self.total)

    def reset(self):
        self.correct = 0
        self.total = 0


def eval_acc(model: torch.nn.Module, loader, device,
             max_batches: Optional[int] = None) -> float:
    acc = Accuracy()
    model.eval()
    with torch.no_grad():
        for i, (x, y) in enumerate(loader):
            if max_batches is not None and i >= max_batches:
                break
            out = model(x.to(device))
            acc.update(y.to(device), out)
    model.train()
    return acc.get()
