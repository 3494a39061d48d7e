#!/usr/bin/env python3
# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""Decentralized MNIST-style training (reference analog:
examples/pytorch_mnist.py). This environment has no network access, so the
loader falls back to a synthetic MNIST-shaped dataset when the real one is
absent; pass --data-dir to use downloaded MNIST tensors if available.

    ./bfrun -np 4 python examples/pytorch_mnist.py --epochs 2
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


import argparse
import os

import torch
import torch.nn as nn
import torch.nn.functional as F

import bluefog_amd as bf
import bluefog_amd.parallel.topology as tu


class Net(nn.Module):
    def __init__(self):
        super().__init__()
        self.conv1 = nn.Conv2d(1, 32, 3, 1)
        self.conv2 = nn.Conv2d(32, 64, 3, 1)
        self.fc1 = nn.Linear(9216, 128)
        self.fc2 = nn.Linear(128, 10)

    def forward(self, x):
        x = F.relu(self.conv1(x))
        x = F.max_pool2d(F.relu(self.conv2(x)), 2)
        x = torch.flatten(x, 1)
        return self.fc2(F.relu(self.fc1(x)))


def synthetic_mnist(n, seed):
    # class-conditional Gaussian blobs so the problem is actually learnable;
    # the class CENTERS are shared across ranks (fixed seed) so every
    # rank's labels mean the same thing — only the samples are per-rank
    gc = torch.Generator().manual_seed(4242)
    centers = torch.randn(10, 1, 28, 28, generator=gc)
    g = torch.Generator().manual_seed(seed)
    y = torch.randint(0, 10, (n,), generator=g)
    x = centers[y] + 0.5 * torch.randn(n, 1, 28, 28, generator=g)
    return torch.utils.data.TensorDataset(x, y)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--epochs", type=int, default=2)
    p.add_argument("--batch-size", type=int, default=64)
    p.add_argument("--lr", type=float, default=0.01)
    p.add_argument("--samples-per-rank", type=int, default=4096)
    p.add_argument("--dist-optimizer", default="neighbor_allreduce",
                   choices=["neighbor_allreduce", "gradient_allreduce", "win_put"])
    args = p.parse_args()

    bf.init()
    device = torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")
    topo = bf.ExponentialTwoGraph(bf.size())
    bf.set_topology(topo)

    dataset = synthetic_mnist(args.samples_per_rank, seed=1000 + bf.rank())
    loader = torch.utils.data.DataLoader(dataset, batch_size=args.batch_size, shuffle=True)

    torch.manual_seed(0)
    model = Net().to(device)
    base = torch.optim.SGD(model.parameters(), lr=args.lr, momentum=0.9)
    if args.dist_optimizer == "gradient_allreduce":
        optimizer = bf.DistributedGradientAllreduceOptimizer(base, model=model)
    elif args.dist_optimizer == "win_put":
        optimizer = bf.DistributedWinPutOptimizer(base, model=model)
    else:
        optimizer = bf.DistributedAdaptWithCombineOptimizer(
            base, model=model,
            communication_type=bf.CommunicationType.neighbor_allreduce,
        )
    bf.broadcast_parameters(model.state_dict(), root_rank=0)
    bf.broadcast_optimizer_state(base, root_rank=0)

    dyn = (
        tu.GetDynamicOnePeerSendRecvRanks(topo, bf.rank())
        if bf.size() > 1 and args.dist_optimizer == "neighbor_allreduce"
        else None
    )
    model.train()
    for epoch in range(args.epochs):
        total, correct, loss_sum = 0, 0, 0.0
        for x, y in loader:
            x, y = x.to(device), y.to(device)
            if dyn is not None:
                send, recv = next(dyn)
                w = 1.0 / (len(recv) + 1)
                optimizer.self_weight = w
                optimizer.src_weights = {r: w for r in recv}
                optimizer.dst_weights = send
            optimizer.zero_grad()
            out = model(x)
            loss = F.cross_entropy(out, y)
            loss.backward()
            optimizer.step()
            loss_sum += loss.item() * y.numel()
            correct += (out.argmax(1) == y).sum().item()
            total += y.numel()
        stats = bf.allreduce(
            torch.tensor([loss_sum, correct, total], dtype=torch.float64), average=False
        )
        if bf.rank() == 0:
            print(
                f"epoch {epoch}: loss {stats[0] / stats[2]:.4f}, "
                f"acc {stats[1] / stats[2]:.4f}"
            )


if __name__ == "__main__":
    main()
