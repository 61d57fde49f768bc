#!/usr/bin/env python3
"""Bi-Sparse + DGT composed training (BASELINE config 5 composition):
the kvstore ships the Bi-Sparse payload with DGT's 4-bit tier over the
packed values (the reference's ENABLE_DGT applies to every push,
including BSC-compressed ones — kv_app.h:917-995). Update-on-worker
like cnn_bsc.py (examples/cnn_bsc.py:77-122 structure)."""

import torch

from common import base_parser, setup, train_loop


def main():
    p = base_parser()
    p.add_argument("--bsc-ratio", type=float, default=0.01)
    args = p.parse_args()

    kv, net, device = setup(args, mode="dist_sync",
                            dgt_block_size=1024, dgt_k=0.5)
    kv.set_gradient_compression({"type": "bsc_dgt",
                                 "threshold": args.bsc_ratio})
    opt = torch.optim.SGD(net.parameters(), lr=args.learning_rate)

    def step(params, num_samples):
        for idx, q in enumerate(params):
            kv.push(idx, q.grad, priority=-idx)
        for idx, q in enumerate(params):
            g = torch.empty_like(q.grad)
            kv.pull(idx, g, priority=-idx)
            q.grad.copy_(g / num_samples)
        opt.step()

    train_loop(args, kv, net, device, step, tag="bsc_dgt")


if __name__ == "__main__":
    main()
