#!/usr/bin/env python3
# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""Average consensus — the hello-world of decentralized communication and
BASELINE config 1 (CPU/gloo world_size=2 static ring works; so does 8x
MI355X). Every rank holds a random vector; repeated neighbor averaging
drives every rank to the global mean.

    ./bfrun -np 2 python examples/pytorch_average_consensus.py
    ./bfrun -np 2 python examples/pytorch_average_consensus.py --asynchronous-mode
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


import argparse

import torch

import bluefog_amd as bf


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--data-size", type=int, default=100000)
    parser.add_argument("--max-iters", type=int, default=200)
    parser.add_argument("--atol", type=float, default=1e-6)
    parser.add_argument("--asynchronous-mode", action="store_true",
                        help="use one-sided win_put/win_update instead of "
                             "synchronous neighbor_allreduce")
    parser.add_argument("--seed", type=int, default=2026)
    args = parser.parse_args()

    bf.init()
    bf.set_topology(bf.RingGraph(bf.size()))
    device = torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")

    torch.manual_seed(args.seed * (bf.rank() + 1))
    x = torch.randn(args.data_size, device=device)
    x_global_mean = bf.allreduce(x, average=True)

    if not args.asynchronous_mode:
        for it in range(args.max_iters):
            x = bf.neighbor_allreduce(x, name="consensus")
            err = (x - x_global_mean).norm() / max(x_global_mean.norm(), 1e-12)
            if err < args.atol:
                break
    else:
        bf.win_create(x, "consensus_win")
        for it in range(args.max_iters):
            bf.win_put(x, "consensus_win")
            bf.barrier()
            x = bf.win_update("consensus_win")
            err = (x - x_global_mean).norm() / max(x_global_mean.norm(), 1e-12)
            if err < args.atol:
                break
        bf.win_free("consensus_win")

    status = "consensus reached" if err < args.atol else "max iterations hit"
    print(
        f"[rank {bf.rank()}] {status} after {it + 1} iterations, "
        f"relative error {err.item():.3e}"
    )
    assert err < 1e-3, "consensus failed to converge"


if __name__ == "__main__":
    main()
