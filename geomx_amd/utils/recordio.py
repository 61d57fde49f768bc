"""Packed-record dataset files: the im2rec / RecordIO analog.

The reference ships MXNet's RecordIO tooling (tools/im2rec.{py,cc},
stock dmlc-core RecordIO: magic-framed variable-length records read
sequentially or via a .idx offset table). Workers there stream their
shard from a .rec file instead of loose files.

MI355X-native replacement: one flat data file plus a numpy offset
index, memory-mapped at read time. A record is

    [u32 magic][u32 flag][u64 payload_len][payload bytes]

where the payload is a self-describing tensor blob (dtype tag, ndim,
shape, raw bytes) plus an int64 label. Reads are zero-copy slices of
the mmap (torch.frombuffer), so a DataLoader worker touches only the
pages it reads — the right layout for feeding 8 concurrent ranks from
one node-local file. The .idx sidecar makes random access O(1), which
is what SplitSampler / ClassSplitSampler need.
"""

from __future__ import annotations

import mmap
import os
import struct
from typing import Iterable, Optional, Tuple

import numpy as np
import torch
from torch.utils.data import Dataset

try:  # native reader (csrc/recordio.cpp): mmap views + threaded batch
    from .. import _geoio
except ImportError:  # pragma: no cover - built by setup.py
    _geoio = None

_MAGIC = 0xCED7EC0D
_HEADER = struct.Struct("<IIQ")          # magic, flag, payload_len
_DTYPES = {
    0: torch.uint8, 1: torch.int8, 2: torch.int16, 3: torch.int32,
    4: torch.int64, 5: torch.float16, 6: torch.bfloat16, 7: torch.float32,
    8: torch.float64,
}
_DTYPE_TAGS = {v: k for k, v in _DTYPES.items()}


def _pack_tensor(t: torch.Tensor, label: int) -> bytes:
    t = t.detach().contiguous().cpu()
    if t.dtype not in _DTYPE_TAGS:
        raise TypeError(f"unsupported dtype {t.dtype}")
    shape = list(t.shape)
    head = struct.pack("<qBB", int(label), _DTYPE_TAGS[t.dtype], len(shape))
    head += struct.pack(f"<{len(shape)}q", *shape) if shape else b""
    if t.dtype == torch.bfloat16:
        raw = t.view(torch.uint16).numpy().tobytes()
    else:
        raw = t.numpy().tobytes()
    return head + raw


def _unpack_tensor(buf: memoryview) -> Tuple[torch.Tensor, int]:
    label, tag, ndim = struct.unpack_from("<qBB", buf, 0)
    off = 10
    shape = struct.unpack_from(f"<{ndim}q", buf, off) if ndim else ()
    off += 8 * ndim
    dtype = _DTYPES[tag]
    numel = 1
    for s in shape:
        numel *= s
    if dtype == torch.bfloat16:
        t = torch.frombuffer(buf, dtype=torch.uint16, count=numel,
                             offset=off).view(torch.bfloat16)
    else:
        t = torch.frombuffer(buf, dtype=dtype, count=numel, offset=off)
    return t.reshape(shape).clone(), label


class RecordWriter:
    """Append-only writer for `<path>` + `<path>.idx`."""

    def __init__(self, path: str):
        self.path = path
        self._f = open(path, "wb")
        self._offsets = []

    def write(self, tensor: torch.Tensor, label: int = 0, flag: int = 0):
        payload = _pack_tensor(tensor, label)
        self._offsets.append(self._f.tell())
        self._f.write(_HEADER.pack(_MAGIC, flag, len(payload)))
        self._f.write(payload)

    def close(self):
        if self._f is None:
            return
        self._f.close()
        self._f = None
        np.asarray(self._offsets, dtype=np.int64).tofile(self.path + ".idx")

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.close()


def pack_dataset(dataset, path: str,
                 indices: Optional[Iterable[int]] = None) -> int:
    """im2rec analog: pack any (tensor, label) map-style dataset into a
    record file. Returns the number of records written."""
    n = 0
    with RecordWriter(path) as w:
        for i in (indices if indices is not None else range(len(dataset))):
            x, y = dataset[i]
            w.write(x, int(y))
            n += 1
    return n


class RecordDataset(Dataset):
    """Random-access reader; safe to use from forked/spawned DataLoader
    workers (each lazily opens its own map). Reads go through the
    native C++ reader (`geomx_amd._geoio`, csrc/recordio.cpp — the
    reference's C++ IO path analog) when it is built; `native=False`
    forces the pure-Python mmap fallback. `read_batch` assembles a
    whole same-shape batch in one call with a multi-threaded copy."""

    def __init__(self, path: str, native: Optional[bool] = None):
        self.path = path
        idx_path = path + ".idx"
        if not os.path.exists(idx_path):
            raise FileNotFoundError(idx_path)
        self.offsets = np.fromfile(idx_path, dtype=np.int64)
        self._mm = None
        self.native = (_geoio is not None) if native is None else native
        if self.native and _geoio is None:
            raise RuntimeError("native reader requested but _geoio "
                               "extension is not built")
        self._nf = None

    def _native_file(self):
        if self._nf is None:
            self._nf = _geoio.RecordFile(self.path)
        return self._nf

    def read_batch(self, indices, threads: int = 4,
                   pin_memory: bool = False):
        """Assemble records `indices` (same shape/dtype) into one
        [N, *shape] tensor + [N] int64 labels."""
        idx = [int(i) for i in indices]
        if self.native:
            return self._native_file().read_batch(idx, threads,
                                                  pin_memory)
        xs, ys = zip(*(self[i] for i in idx))
        batch = torch.stack(xs)
        if pin_memory:
            batch = batch.pin_memory()
        return batch, torch.tensor(ys, dtype=torch.int64)

    def _map(self) -> mmap.mmap:
        if self._mm is None:
            f = open(self.path, "rb")
            self._mm = mmap.mmap(f.fileno(), 0, access=mmap.ACCESS_READ)
            f.close()
        return self._mm

    def __len__(self):
        return len(self.offsets)

    def __getitem__(self, i: int):
        if self.native:
            return self._native_file().read(int(i))
        mm = self._map()
        off = int(self.offsets[i])
        magic, _flag, plen = _HEADER.unpack_from(mm, off)
        if magic != _MAGIC:
            raise IOError(f"corrupt record at offset {off} in {self.path}")
        start = off + _HEADER.size
        return _unpack_tensor(memoryview(mm)[start:start + plen])

    @property
    def labels(self) -> torch.Tensor:
        """All labels (one header read per record) — feeds
        ClassSplitSampler without materializing the tensors."""
        if self.native:
            return self._native_file().labels()
        mm = self._map()
        out = torch.empty(len(self), dtype=torch.int64)
        for i, off in enumerate(self.offsets):
            start = int(off) + _HEADER.size
            out[i] = struct.unpack_from("<q", mm, start)[0]
        return out

    # pickling (DataLoader spawn workers): drop the handles, reopen
    # lazily in the worker
    def __getstate__(self):
        d = dict(self.__dict__)
        d["_mm"] = None
        d["_nf"] = None
        return d


class RecordBatchLoader:
    """Iterable of (batch, labels) from a record file — the fast host
    path for feeding one training rank from a node-local shard.
    Uses `RecordDataset.read_batch` (native threaded assembly when
    `_geoio` is built), so one call produces the whole [B, *shape]
    tensor instead of B python-object round-trips. `sampler` orders the
    epoch (e.g. SplitSampler / ClassSplitSampler for IID / non-IID
    sharding); default is sequential."""

    def __init__(self, dataset: "RecordDataset", batch_size: int,
                 sampler=None, threads: int = 4, pin_memory: bool = False,
                 drop_last: bool = False):
        self.ds = dataset
        self.bs = int(batch_size)
        self.sampler = sampler
        self.threads = threads
        self.pin_memory = pin_memory
        self.drop_last = drop_last

    def _indices(self):
        if self.sampler is not None:
            return [int(i) for i in self.sampler]
        return list(range(len(self.ds)))

    def __len__(self):
        n = len(self.sampler) if self.sampler is not None else len(self.ds)
        return n // self.bs if self.drop_last else (n + self.bs - 1) // self.bs

    def __iter__(self):
        idx = self._indices()
        for lo in range(0, len(idx), self.bs):
            chunk = idx[lo:lo + self.bs]
            if self.drop_last and len(chunk) < self.bs:
                return
            yield self.ds.read_batch(chunk, threads=self.threads,
                                     pin_memory=self.pin_memory)
