#!/usr/bin/env python3
"""FP16-transmission training (reference examples/cnn_fp16.py):
update-on-server with fp16 wire format on the WAN tier and an fp32
master copy at the server (multi_precision, kvstore_dist_server.h:
374-407)."""

from common import base_parser, setup, train_loop

from geomx_amd.kvstore.optimizer import OptimizerSpec


def main():
    args = base_parser().parse_args()
    kv, net, device = setup(args)
    kv.set_gradient_compression({"type": "fp16"})
    kv.set_optimizer(OptimizerSpec("adam", lr=args.learning_rate))

    def step(params, num_samples):
        for idx, p_ in enumerate(params):
            kv.push(idx, p_.grad / num_samples, priority=-idx)
            kv.pull(idx, p_.data, priority=-idx)

    train_loop(args, kv, net, device, step, tag="fp16")


if __name__ == "__main__":
    main()
