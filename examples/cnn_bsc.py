#!/usr/bin/env python3
"""Bi-Sparse compressed training (reference examples/cnn_bsc.py):
update-on-worker. Workers push grad/num_samples; the WAN tier exchanges
Bi-Sparse compressed sums; workers pull the AGGREGATED GRADIENT and run
a local Adam step (Trainer update_on_kvstore=False analog)."""

import torch

from common import base_parser, setup, train_loop

from geomx_amd.kvstore.optimizer import OptimizerSpec, ServerOptimizer


def main():
    p = base_parser()
    p.add_argument("-bcr", "--bisparse-compression-ratio", type=float,
                   default=0.01)
    args = p.parse_args()
    assert 0 < args.bisparse_compression_ratio < 1

    kv, net, device = setup(args)
    kv.set_gradient_compression({"type": "bsc",
                                 "threshold": args.bisparse_compression_ratio})
    local_opt = ServerOptimizer(OptimizerSpec("adam", lr=args.learning_rate))

    def step(params, num_samples):
        grads = []
        for idx, p_ in enumerate(params):
            kv.push(idx, p_.grad / num_samples, priority=-idx)
            g = torch.empty_like(p_.grad)
            kv.pull(idx, g, priority=-idx)
            grads.append(g)
        with torch.no_grad():
            for idx, (p_, g) in enumerate(zip(params, grads)):
                local_opt.update(idx, p_.data.reshape(-1), g.reshape(-1),
                                 rescale=1.0 / kv.num_all_workers)

    train_loop(args, kv, net, device, step, tag="bsc")


if __name__ == "__main__":
    main()
