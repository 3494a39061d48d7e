#!/usr/bin/env python3
# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""Classic decentralized optimization algorithms on a synthetic logistic
regression problem (reference analog: examples/pytorch_optimization.py —
diffusion, exact diffusion, gradient tracking, push-DIGing; the algorithms
are the published ones, implemented here from their update equations).

    ./bfrun -np 4 python examples/pytorch_optimization.py --method diffusion
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


import argparse

import torch

import bluefog_amd as bf


def make_problem(rank: int, n=64, d=32, seed=2026):
    g = torch.Generator().manual_seed(seed)
    w_star = torch.randn(d, 1, generator=g)
    g2 = torch.Generator().manual_seed(seed + 1 + rank)
    A = torch.randn(n, d, generator=g2)
    y = torch.sign(A @ w_star + 0.1 * torch.randn(n, 1, generator=g2))
    rho = 1e-2
    return A, y, rho


def grad_fn(A, y, rho, w):
    """Gradient of l2-regularized logistic loss (mean over local samples)."""
    z = A @ w
    return -(A.t() @ (y / (1 + torch.exp(y * z)))) / A.shape[0] + rho * w


def global_grad_norm(A, y, rho, w):
    g = grad_fn(A, y, rho, w)
    return bf.allreduce(g, average=True).norm().item()


def diffusion(A, y, rho, w, lr, iters):
    """Adapt-then-combine diffusion: phi = w - lr*grad; w = neighbor_avg(phi)."""
    for _ in range(iters):
        phi = w - lr * grad_fn(A, y, rho, w)
        w = bf.neighbor_allreduce(phi, name="diffusion.w")
    return w

def exact_diffusion(A, y, rho, w, lr, iters):
    """Exact diffusion / D2: psi = w - lr*grad; phi = psi + w - psi_prev;
    w = neighbor_avg(phi). Removes the steady-state bias of diffusion."""
    psi_prev = w.clone()
    for _ in range(iters):
        psi = w - lr * grad_fn(A, y, rho, w)
        phi = psi + w - psi_prev
        w = bf.neighbor_allreduce(phi, name="exact_diffusion.phi")
        psi_prev = psi
    return w


def gradient_tracking(A, y, rho, w, lr, iters):
    """DSGT: maintain a tracker q of the global gradient:
    w+ = neighbor_avg(w) - lr*q;  q+ = neighbor_avg(q) + grad(w+) - grad(w)."""
    q = grad_fn(A, y, rho, w)
    g_prev = q.clone()
    for _ in range(iters):
        w_next = bf.neighbor_allreduce(w, name="gt.w") - lr * q
        g_next = grad_fn(A, y, rho, w_next)
        q = bf.neighbor_allreduce(q, name="gt.q") + g_next - g_prev
        w, g_prev = w_next, g_next
    return w


def push_diging(A, y, rho, w, lr, iters):
    """Push-DIGing over a directed exp2 graph with column-stochastic
    weights, implemented on one-sided win_accumulate (reference analog:
    pytorch_optimization.py:371-435)."""
    outdeg = len(bf.out_neighbor_ranks())
    self_w = 1.0 / (outdeg + 1)
    dst_weights = {r: self_w for r in bf.out_neighbor_ranks()}

    d = w.shape[0]
    x = torch.cat([w, grad_fn(A, y, rho, w), torch.ones(1, 1)], dim=0)
    bf.win_create(x, "push_diging", zero_init=True)
    g_prev = x[d : 2 * d].clone()
    for _ in range(iters):
        x[:d] -= lr * x[d : 2 * d]
        bf.win_accumulate(x, "push_diging", dst_weights=dst_weights, require_mutex=True)
        x.mul_(self_w)
        bf.barrier()
        x = bf.win_update_then_collect("push_diging")
        u = x[:d] / x[-1]
        g_new = grad_fn(A, y, rho, u)
        x[d : 2 * d] += g_new - g_prev
        g_prev = g_new
    bf.barrier()
    bf.win_free("push_diging")
    return x[:d] / x[-1]


METHODS = {
    "diffusion": diffusion,
    "exact_diffusion": exact_diffusion,
    "gradient_tracking": gradient_tracking,
    "push_diging": push_diging,
}


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--method", default="diffusion", choices=sorted(METHODS))
    p.add_argument("--lr", type=float, default=0.5)
    p.add_argument("--iters", type=int, default=300)
    args = p.parse_args()

    bf.init()
    bf.set_topology(bf.ExponentialTwoGraph(bf.size()))
    A, y, rho = make_problem(bf.rank())
    w = torch.zeros(A.shape[1], 1)
    w = METHODS[args.method](A, y, rho, w, args.lr, args.iters)
    # consensus + optimality check on the averaged iterate
    w_avg = bf.allreduce(w, average=True)
    gnorm = global_grad_norm(A, y, rho, w_avg)
    consensus = (w - w_avg).norm().item()
    if bf.rank() == 0:
        print(
            f"{args.method}: global grad norm {gnorm:.3e}, "
            f"consensus error {consensus:.3e}"
        )


if __name__ == "__main__":
    main()
