#!/usr/bin/env python3
# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""Standalone one-sided-window latency/throughput probe.

Dev-tool analog of the reference's standalone MPI experiments
(scripts/mpi_win_ops.cc, scripts/mpi_passive_recv.cc — not shipped,
used to characterize the window data plane). This probe measures, per
payload size:

- win_put latency (origin-side, blocking) and the store round-trips it
  costs (ControlStore.rpc_counts);
- win_update (fold + ack) latency on the destination;
- effective one-sided bandwidth.

Self-spawns two ranks when run without a rendezvous env:

    python scripts/win_latency_bench.py [--sizes 4096,1048576,...]
    python scripts/win_latency_bench.py --device cuda   # on a GPU box
"""

import argparse
import json
import os
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def parse_args():
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--sizes", default="1024,65536,1048576,16777216",
                   help="comma-separated element counts (fp32)")
    p.add_argument("--iters", type=int, default=50)
    p.add_argument("--device", default="cpu", choices=["cpu", "cuda"])
    return p.parse_args()


def main():
    args = parse_args()
    if "WORLD_SIZE" not in os.environ:
        from bluefog_amd.run import bfrun

        sys.exit(bfrun.main(["-np", "2", "--", sys.executable,
                             os.path.abspath(__file__)] + sys.argv[1:]))

    import torch

    import bluefog_amd as bf

    bf.init()
    assert bf.size() == 2, "this probe runs with exactly 2 ranks"
    bf.set_topology(bf.RingGraph(2))
    dev = torch.device(args.device)
    rank = bf.rank()
    store = bf._ctx().store
    rows = []
    for numel in (int(s) for s in args.sizes.split(",")):
        t = torch.zeros(numel, device=dev)
        name = f"probe{numel}"
        bf.win_create(t, name, zero_init=True)
        bf.barrier()
        peer = 1 - rank
        # warmup
        for _ in range(5):
            bf.win_put(torch.ones(numel, device=dev), name,
                       dst_weights={peer: 1.0})
            bf.win_update(name)
        bf.barrier()
        rpc0 = sum(store.rpc_counts.values())
        src = torch.full((numel,), float(rank + 1), device=dev)
        t0 = time.perf_counter()
        for _ in range(args.iters):
            bf.win_put(src, name, dst_weights={peer: 1.0})
        if dev.type == "cuda":
            torch.cuda.synchronize()
        put_s = (time.perf_counter() - t0) / args.iters
        rpcs = (sum(store.rpc_counts.values()) - rpc0) / args.iters
        t0 = time.perf_counter()
        for _ in range(args.iters):
            bf.win_update(name)
        if dev.type == "cuda":
            torch.cuda.synchronize()
        upd_s = (time.perf_counter() - t0) / args.iters
        bf.barrier()
        bf.win_free(name)
        rows.append({
            "numel": numel,
            "put_us": put_s * 1e6,
            "put_GBps": numel * 4 / put_s / 1e9,
            "put_store_rpcs": rpcs,
            "update_us": upd_s * 1e6,
        })
    if rank == 0:
        print(json.dumps(rows, indent=1), flush=True)
    bf.shutdown()


if __name__ == "__main__":
    main()
