#!/usr/bin/env python3
"""Mixed-precision quantization (reference examples/cnn_mpq.py):
small tensors travel fp16, large tensors Bi-Sparse
(MXNET_KVSTORE_SIZE_LOWER_BOUND gate). Update-on-worker."""

import torch

from common import base_parser, setup, train_loop

from geomx_amd.kvstore.optimizer import OptimizerSpec, ServerOptimizer


def main():
    p = base_parser()
    p.add_argument("-bcr", "--bisparse-compression-ratio", type=float,
                   default=0.01)
    p.add_argument("--size-lower-bound", type=int, default=200000)
    args = p.parse_args()

    kv, net, device = setup(args)
    kv.set_gradient_compression({
        "type": "mpq",
        "threshold": args.bisparse_compression_ratio,
        "size_lower_bound": args.size_lower_bound,
    })
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

    train_loop(args, kv, net, device, step, tag="mpq")


if __name__ == "__main__":
    main()
