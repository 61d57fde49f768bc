#!/usr/bin/env python3
"""ResNet-50 HiPS training (BASELINE.json config 5): the larger model
family through the high-performance GeoTrainer path (bucketed,
backward-overlapped) instead of per-parameter push/pull — the
recommended pattern for real models. Supports the same HiPS knobs as
bench.py (--parties via torchrun world split, compression, WAN cap via
GEOMX_* env)."""

import torch

from common import base_parser, topo_world, make_loaders, eval_acc

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from geomx_amd import Config  # noqa: E402
from geomx_amd.kvstore.optimizer import OptimizerSpec  # noqa: E402
from geomx_amd.models import create_model  # noqa: E402
from geomx_amd.parallel import GeoTrainer  # noqa: E402
from geomx_amd.topology import init_topology  # noqa: E402


def main():
    p = base_parser()
    p.add_argument("--compress", type=str, default=None,
                   choices=[None, "bsc", "fp16", "mpq", "dgt"])
    args = p.parse_args()

    parties = args.parties if topo_world() > 1 else 1
    cfg = Config.from_env(num_parties=parties, compression=args.compress)
    topo = init_topology(parties)
    device = topo.device
    torch.manual_seed(0)
    net = create_model("resnet50", num_classes=10).to(device)
    trainer = GeoTrainer(net, cfg, topo,
                         OptimizerSpec("sgd_mom", lr=args.learning_rate),
                         mode="hips" if parties > 1 else "flat")

    class _KV:  # make_loaders only needs sharding info
        num_all_workers = topo.num_all_workers
        rank = topo.rank
    train_iter, test_iter = make_loaders(args, _KV, device)

    begin = time.time()
    it = 0
    for epoch in range(args.epoch):
        for x, y in train_iter:
            x, y = x.to(device), y.to(device)
            loss = torch.nn.functional.cross_entropy(net(x), y)
            trainer.zero_grad()
            loss.backward()
            trainer.step()
            it += 1
            if topo.rank == 0:
                acc = eval_acc(net, test_iter, device, max_batches=1)
                print("[resnet50][Time %.3f][Epoch %d][Iteration %d] "
                      "Test Acc %.4f"
                      % (time.time() - begin, epoch, it, acc), flush=True)
            if args.max_iters and it >= args.max_iters:
                return


if __name__ == "__main__":
    main()
