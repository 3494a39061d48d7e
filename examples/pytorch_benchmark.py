#!/usr/bin/env python3
# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""Synthetic-data training throughput benchmark — the reference's headline
harness re-built for MI355X (reference: examples/pytorch_benchmark.py:
img/sec = batch_size x batches / wall time, 10 warmup batches, 10x10 timed
batches, per-rank numbers summed by allreduce).

    ./bfrun -np 8 python examples/pytorch_benchmark.py \
        --model resnet50 --batch-size 64 --dist-optimizer neighbor_allreduce
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


import argparse
import time

import numpy as np
import torch

import bluefog_amd as bf
import bluefog_amd.parallel.topology as tu


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="resnet50")
    p.add_argument("--batch-size", type=int, default=64)
    p.add_argument("--num-warmup-batches", type=int, default=10)
    p.add_argument("--num-batches-per-iter", type=int, default=10)
    p.add_argument("--num-iters", type=int, default=10)
    p.add_argument(
        "--dist-optimizer",
        default="neighbor_allreduce",
        choices=[
            "neighbor_allreduce",
            "hierarchical_neighbor_allreduce",
            "gradient_allreduce",
            "allreduce",
            "win_put",
            "pushsum",
            "horovod",  # alias of gradient_allreduce, reference flag parity
        ],
    )
    p.add_argument("--atc-style", action="store_true",
                   help="use DistributedAdaptThenCombineOptimizer")
    p.add_argument("--num-classes", type=int, default=1000)
    p.add_argument("--profiler", action="store_true",
                   help="trace one timed iteration with torch.profiler")
    p.add_argument("--disable-dynamic-topology", action="store_true")
    p.add_argument("--no-cuda", action="store_true")
    args = p.parse_args()

    bf.init()
    device = (
        torch.device("cuda")
        if torch.cuda.is_available() and not args.no_cuda
        else torch.device("cpu")
    )
    if device.type == "cuda":
        torch.backends.cudnn.benchmark = True

    from bluefog_amd import models

    model = getattr(models, args.model)(num_classes=args.num_classes).to(device)
    if device.type == "cuda":
        model = model.to(memory_format=torch.channels_last)

    base_opt = torch.optim.SGD(model.parameters(), lr=0.01, momentum=0.9)
    topo = bf.ExponentialTwoGraph(bf.size())
    bf.set_topology(topo)

    name = args.dist_optimizer
    awc_or_atc = (
        bf.DistributedAdaptThenCombineOptimizer
        if args.atc_style
        else bf.DistributedAdaptWithCombineOptimizer
    )
    if name in ("gradient_allreduce", "horovod"):
        optimizer = bf.DistributedGradientAllreduceOptimizer(base_opt, model=model)
    elif name == "win_put":
        optimizer = bf.DistributedWinPutOptimizer(base_opt, model=model)
    elif name == "pushsum":
        optimizer = bf.DistributedPushSumOptimizer(base_opt, model=model)
    elif name == "hierarchical_neighbor_allreduce":
        bf.set_machine_topology(bf.ExponentialTwoGraph(bf.machine_size()))
        optimizer = awc_or_atc(
            base_opt, model=model,
            communication_type=bf.CommunicationType.hierarchical_neighbor_allreduce,
        )
    else:
        ct = (
            bf.CommunicationType.allreduce
            if name == "allreduce"
            else bf.CommunicationType.neighbor_allreduce
        )
        optimizer = awc_or_atc(base_opt, model=model, communication_type=ct)

    # per-iteration dynamic schedules (reference pytorch_benchmark.py:160-200)
    dyn_gen = None
    dyn_machine_gen = None
    if not args.disable_dynamic_topology and bf.size() > 1:
        if name == "neighbor_allreduce":
            if bf.is_homogeneous() and bf.size() > bf.local_size():
                dyn_gen = tu.GetInnerOuterExpo2DynamicSendRecvRanks(
                    bf.size(), local_size=bf.local_size(), self_rank=bf.rank()
                )
            else:
                dyn_gen = tu.GetDynamicOnePeerSendRecvRanks(topo, bf.rank())
        elif name == "hierarchical_neighbor_allreduce" and bf.machine_size() > 1:
            dyn_machine_gen = tu.GetExp2DynamicSendRecvMachineRanks(
                world_size=bf.size(), local_size=bf.local_size(),
                self_rank=bf.rank(), local_rank=bf.local_rank(),
            )

    bf.broadcast_parameters(model.state_dict(), root_rank=0)

    data = torch.randn(args.batch_size, 3, 224, 224).to(device)
    target = torch.randint(0, 1000, (args.batch_size,)).to(device)
    if device.type == "cuda":
        data = data.to(memory_format=torch.channels_last)
    loss_fn = torch.nn.CrossEntropyLoss()

    step_counter = [0]

    def benchmark_step():
        it = step_counter[0]
        step_counter[0] += 1
        if dyn_gen is not None:
            send, recv = next(dyn_gen)
            w = 1.0 / (len(recv) + 1)
            optimizer.self_weight = w
            optimizer.src_weights = {r: w for r in recv}
            optimizer.dst_weights = send
            optimizer.enable_topo_check = False
        elif dyn_machine_gen is not None:
            send_m, recv_m = next(dyn_machine_gen)
            w = 1.0 / (len(recv_m) + 1)
            optimizer.self_weight = w
            optimizer.src_machine_weights = {r: w for r in recv_m}
            optimizer.dst_machine_weights = send_m
            optimizer.enable_topo_check = False
        elif (
            name == "win_put"
            and not args.disable_dynamic_topology
            and bf.out_neighbor_ranks()
        ):
            # rotate the gossip destination, one out-neighbor per iteration
            nbrs = bf.out_neighbor_ranks()
            optimizer.dst_weights = {nbrs[it % len(nbrs)]: 1.0}
        optimizer.zero_grad()
        loss = loss_fn(model(data), target)
        loss.backward()
        optimizer.step()

    for _ in range(args.num_warmup_batches):
        benchmark_step()

    if args.profiler:
        # reference --profiler used torch.autograd.profiler around one
        # iteration; torch.profiler is its successor
        from torch.profiler import ProfilerActivity, profile

        acts = [ProfilerActivity.CPU]
        if device.type == "cuda":
            acts.append(ProfilerActivity.CUDA)
        with profile(activities=acts) as prof:
            for _ in range(args.num_batches_per_iter):
                benchmark_step()
            if device.type == "cuda":
                torch.cuda.synchronize()
        if bf.rank() == 0:
            print(prof.key_averages().table(sort_by="self_cuda_time_total"
                                            if device.type == "cuda"
                                            else "self_cpu_time_total",
                                            row_limit=25))

    img_secs = []
    for _ in range(args.num_iters):
        if device.type == "cuda":
            torch.cuda.synchronize()
        t0 = time.time()
        for _ in range(args.num_batches_per_iter):
            benchmark_step()
        if device.type == "cuda":
            torch.cuda.synchronize()
        img_sec = args.batch_size * args.num_batches_per_iter / (time.time() - t0)
        img_secs.append(img_sec)

    img_sec_mean = np.mean(img_secs)
    img_sec_conf = 1.96 * np.std(img_secs)
    print(f"[rank {bf.rank()}] Img/sec per GPU: {img_sec_mean:.1f} +- {img_sec_conf:.1f}")
    total = bf.allreduce(torch.tensor([img_sec_mean]), average=False)
    if bf.rank() == 0:
        print(
            f"Total img/sec on {bf.size()} GPU(s): "
            f"{total.item():.1f} ({args.model}, bs={args.batch_size}, "
            f"{args.dist_optimizer})"
        )


if __name__ == "__main__":
    main()
