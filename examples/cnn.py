#!/usr/bin/env python3
"""Vanilla HiPS training (reference examples/cnn.py): update-on-server.

Workers push grad/num_samples; the global tier (party leaders) runs the
pickled-optimizer-equivalent fused Adam; workers pull PARAMETERS.
Supports --mixed-sync (dist_async global tier) and --dcasgd (delay
compensation), mirroring the reference's flags (cnn.py:66-81).
"""

from common import base_parser, setup, train_loop

from geomx_amd.kvstore.optimizer import OptimizerSpec


def main():
    p = base_parser()
    p.add_argument("--mixed-sync", action="store_true")
    p.add_argument("--dcasgd", action="store_true")
    p.add_argument("--async-transport", type=str, default="lockstep",
                   choices=["lockstep", "store"],
                   help="store = true-async parameter server")
    args = p.parse_args()

    mode = "dist_async" if (args.mixed_sync or args.dcasgd) else "dist_sync"
    kv, net, device = setup(args, mode=mode,
                            async_transport=args.async_transport)
    if args.dcasgd:
        kv.set_optimizer(OptimizerSpec("dcasgd", lr=args.learning_rate))
    else:
        kv.set_optimizer(OptimizerSpec("adam", lr=args.learning_rate))

    def step(params, num_samples):
        for idx, p_ in enumerate(params):
            kv.push(idx, p_.grad / num_samples, priority=-idx)
            kv.pull(idx, p_.data, priority=-idx)

    train_loop(args, kv, net, device, step, tag="vanilla" if mode ==
               "dist_sync" else mode)


if __name__ == "__main__":
    main()
