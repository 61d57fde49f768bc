#!/usr/bin/env python3
"""HFA training (reference examples/cnn_hfa.py): workers run K1 LOCAL
optimizer steps, then push PARAMETERS/num_local_workers (model
averaging); leaders forward to the global tier only every K2-th
aggregation, transmitting milestone deltas
(kvstore_dist_server.h:959-972,1324-1343)."""

import torch

from common import base_parser, setup, train_loop

from geomx_amd.kvstore.optimizer import OptimizerSpec, ServerOptimizer


def main():
    p = base_parser()
    p.add_argument("--k1", type=int, default=2)
    p.add_argument("--k2", type=int, default=2)
    args = p.parse_args()

    kv, net, device = setup(args, use_hfa=True, hfa_k1=args.k1,
                            hfa_k2=args.k2)
    local_opt = ServerOptimizer(OptimizerSpec("adam", lr=args.learning_rate))
    state = {"local_iters": 0}

    def step(params, num_samples):
        # local optimizer step every iteration
        with torch.no_grad():
            for idx, p_ in enumerate(params):
                local_opt.update(idx, p_.data.reshape(-1),
                                 p_.grad.reshape(-1),
                                 rescale=1.0 / num_samples)
        state["local_iters"] += 1
        if state["local_iters"] % args.k1 != 0:
            return
        # every K1 local steps: push averaged MODEL, pull averaged model
        nloc = kv.num_workers
        for idx, p_ in enumerate(params):
            kv.push(idx, p_.data / nloc, priority=-idx)
            kv.pull(idx, p_.data, priority=-idx)

    train_loop(args, kv, net, device, step, tag="hfa")


if __name__ == "__main__":
    main()
