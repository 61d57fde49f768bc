#!/usr/bin/env python3
"""DGT training (reference scripts/*/run_dgt.sh + kv_app.h:842-995):
chunk-priority differential transmission on the WAN tier — the top
DMLC_K fraction of 4KB chunks (by EWMA mean|g| contribution) travel
exact, the rest 4-bit quantized with residual feedback."""

from common import base_parser, setup, train_loop

from geomx_amd.kvstore.optimizer import OptimizerSpec


def main():
    p = base_parser()
    p.add_argument("--dgt-k", type=float, default=0.5)
    args = p.parse_args()

    kv, net, device = setup(args, enable_dgt=3, dgt_k=args.dgt_k)
    kv.set_gradient_compression({"type": "dgt"})
    kv.set_optimizer(OptimizerSpec("adam", lr=args.learning_rate))

    def step(params, num_samples):
        for idx, p_ in enumerate(params):
            kv.push(idx, p_.grad / num_samples, priority=-idx)
            kv.pull(idx, p_.data, priority=-idx)

    train_loop(args, kv, net, device, step, tag="dgt")


if __name__ == "__main__":
    main()
