"""Shared scaffolding for the example training drivers.

Mirrors the reference examples' structure (examples/cnn*.py): build the
CNN, create a kvstore, init/push/pull per parameter each iteration, and
print per-iteration test accuracy. Launch with torchrun, e.g.

  python -m torch.distributed.run --nnodes=1 --nproc-per-node 4 \
      --master-addr 127.0.0.1 examples/cnn.py --parties 2 --epoch 1

Runs on CPU (gloo) or GPU (RCCL) alike. Data is synthetic (no network
in this environment) with learnable class structure.
"""

import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from geomx_amd import Config  # noqa: E402
from geomx_amd.kvstore import create  # noqa: E402
from geomx_amd.models import geo_cnn  # noqa: E402
from geomx_amd.topology import init_topology  # noqa: E402
from geomx_amd.utils.data import SyntheticImageDataset, worker_loader  # noqa: E402
from geomx_amd.utils.metrics import Measure, eval_acc  # noqa: E402


def base_parser():
    p = argparse.ArgumentParser()
    p.add_argument("-lr", "--learning-rate", type=float, default=0.01)
    p.add_argument("-bs", "--batch-size", type=int, default=32)
    p.add_argument("-ep", "--epoch", type=int, default=1)
    p.add_argument("-sc", "--split-by-class", action="store_true")
    p.add_argument("--parties", type=int, default=1)
    p.add_argument("--image-size", type=int, default=28)
    p.add_argument("--data-n", type=int, default=1024)
    p.add_argument("--max-iters", type=int, default=0)
    p.add_argument("--measure-out", type=str, default=None)
    p.add_argument("--global-mode", type=str, default="sharded",
                   choices=["sharded", "replicated"],
                   help="global tier: sharded = MultiGPS key owners; "
                        "replicated = TSEngine-style incast-free ring")
    return p


def setup(args, mode="dist_sync", **cfg_overrides):
    parties = args.parties if topo_world() > 1 else 1
    cfg = Config.from_env(num_parties=parties, **cfg_overrides)
    topo = init_topology(parties)
    kv = create(mode, cfg=cfg, topo=topo,
                global_mode=getattr(args, "global_mode", "sharded"))
    device = topo.device
    torch.manual_seed(0)
    net = geo_cnn(in_channels=3, image_size=args.image_size).to(device)
    return kv, net, device


def topo_world():
    return int(os.environ.get("WORLD_SIZE", "1"))


def make_loaders(args, kv, device):
    shape = (3, args.image_size, args.image_size)
    train = SyntheticImageDataset(args.data_n, shape=shape, seed=1)
    test = SyntheticImageDataset(256, shape=shape, seed=2)
    train_iter = worker_loader(train, args.batch_size, kv.num_all_workers,
                               kv.rank, split_by_class=args.split_by_class)
    test_iter = torch.utils.data.DataLoader(test, batch_size=64)
    return train_iter, test_iter


def train_loop(args, kv, net, device, step_fn, tag):
    """Common loop: for each batch call step_fn(params, grads scaled),
    then evaluate + print (reference cnn.py:117-131 structure)."""
    train_iter, test_iter = make_loaders(args, kv, device)
    params = [p for p in net.parameters() if p.requires_grad]
    for idx, p in enumerate(params):
        kv.init(idx, p.data)
        kv.pull(idx, p.data)

    measure = Measure()
    begin = time.time()
    it = 0
    for epoch in range(args.epoch):
        for x, yb in train_iter:
            x, yb = x.to(device), yb.to(device)
            measure.start("compute")
            loss = torch.nn.functional.cross_entropy(net(x), yb)
            net.zero_grad()
            loss.backward()
            measure.stop("compute")
            measure.start("sync")
            step_fn(params, x.shape[0])
            measure.stop("sync")
            it += 1
            acc = eval_acc(net, test_iter, device, max_batches=2)
            if kv.rank == 0:
                print("[%s][Time %.3f][Epoch %d][Iteration %d] Test Acc %.4f"
                      % (tag, time.time() - begin, epoch, it, acc), flush=True)
            measure.next_iteration(iter=it, acc=acc)
            if args.max_iters and it >= args.max_iters:
                break
        if args.max_iters and it >= args.max_iters:
            break
    if args.measure_out and kv.rank == 0:
        measure.dump(args.measure_out)
    return net
