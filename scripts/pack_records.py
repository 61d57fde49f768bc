#!/usr/bin/env python3
"""Record-file packer CLI — the tools/im2rec.py analog.

Packs a dataset into the framework's record-shard format (see
geomx_amd/utils/recordio.py). Sources:

  --synthetic N        pack N synthetic labelled images (the in-repo
                       dataset class; useful for IO benchmarking)
  --tensors FILE.pt    pack a torch-saved (tensor[N,...], labels[N])
                       tuple or {"x": ..., "y": ...} dict

Optionally shard the output per worker (--shards W writes
out.rec.0 .. out.rec.W-1 with contiguous splits — one node-local
shard per training rank).

  python scripts/pack_records.py --synthetic 2048 --image-size 64 \
      --out /tmp/train.rec --shards 8
"""

import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from geomx_amd.utils.data import SyntheticImageDataset  # noqa: E402
from geomx_amd.utils.recordio import RecordDataset, pack_dataset  # noqa: E402


class _TensorPair:
    def __init__(self, x, y):
        assert x.shape[0] == y.shape[0], "x/y length mismatch"
        self.x, self.y = x, y

    def __len__(self):
        return self.x.shape[0]

    def __getitem__(self, i):
        return self.x[i], int(self.y[i])


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--out", required=True, help="output .rec path")
    ap.add_argument("--synthetic", type=int, default=0,
                    help="pack N synthetic images")
    ap.add_argument("--image-size", type=int, default=224)
    ap.add_argument("--channels", type=int, default=3)
    ap.add_argument("--num-classes", type=int, default=10)
    ap.add_argument("--seed", type=int, default=0)
    ap.add_argument("--tensors", type=str, default=None,
                    help="torch.save'd (x, y) tuple or {'x','y'} dict")
    ap.add_argument("--shards", type=int, default=1,
                    help="write W contiguous per-worker shards")
    args = ap.parse_args()

    if args.synthetic:
        ds = SyntheticImageDataset(
            args.synthetic,
            shape=(args.channels, args.image_size, args.image_size),
            num_classes=args.num_classes, seed=args.seed)
    elif args.tensors:
        blob = torch.load(args.tensors, map_location="cpu",
                          weights_only=True)
        if isinstance(blob, dict):
            ds = _TensorPair(blob["x"], blob["y"])
        else:
            ds = _TensorPair(*blob)
    else:
        ap.error("one of --synthetic / --tensors is required")

    n = len(ds)
    if args.shards <= 1:
        written = pack_dataset(ds, args.out)
        print(f"{args.out}: {written} records "
              f"({os.path.getsize(args.out) / 1e6:.1f} MB)")
        RecordDataset(args.out)  # sanity: index readable
        return
    per = (n + args.shards - 1) // args.shards
    for w in range(args.shards):
        path = f"{args.out}.{w}"
        idx = range(w * per, min(n, (w + 1) * per))
        written = pack_dataset(ds, path, indices=idx)
        print(f"{path}: {written} records")
        RecordDataset(path)


if __name__ == "__main__":
    main()
