#!/usr/bin/env python3
# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""Microbenchmark of the bluefog_amd native CDNA4 kernels on one MI355X.

Times each hand-written kernel on synthetic data shaped like the flagship
workload (ResNet50 = 25.56M fp32 params flattened into one bucket; the
dynamic one-peer exp2 configuration gathers 1 neighbor slice, the static
exp2 8-GPU configuration gathers 3) and reports effective HBM bandwidth
(bytes moved / time) against the ~8 TB/s HBM3E peak.

These are the kernels that replace the reference's torch post-op chains
(mpi_ops.cc:99-164 neighbor averaging, optimizers.py:601-760 parameter-wise
steps, cuda_kernels.cu:24-116 buffer scaling); see
bluefog_amd/csrc/bluefog_kernels.hip.

    python bench_kernels.py [--numel N] [--iters K] [--json]
"""

import argparse
import json

import torch

from bluefog_amd import _C
from bluefog_amd.ops import hip_ext


def time_kernel(fn, iters, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    start = torch.cuda.Event(enable_timing=True)
    end = torch.cuda.Event(enable_timing=True)
    start.record()
    for _ in range(iters):
        fn()
    end.record()
    torch.cuda.synchronize()
    return start.elapsed_time(end) / iters  # ms


def run(numel, iters, dtype, dev):
    esz = torch.tensor([], dtype=dtype).element_size()
    rows = []

    def add(name, ms, bytes_moved):
        rows.append(
            {
                "kernel": name,
                "dtype": str(dtype).replace("torch.", ""),
                "numel": numel,
                "ms": ms,
                "GBps": bytes_moved / ms / 1e6,
            }
        )

    self_t = torch.randn(numel, device=dev).to(dtype)
    grad = torch.randn(numel, device=dev).to(dtype)
    out = torch.empty_like(self_t)

    for n_nbr in (1, 3, 7):
        gathered = torch.randn(n_nbr * numel, device=dev).to(dtype)
        w = [1.0 / (n_nbr + 1)] * n_nbr
        sw = 1.0 / (n_nbr + 1)
        ms = time_kernel(
            lambda: hip_ext.weighted_combine(out, self_t, sw, gathered, w), iters
        )
        # reads self + n_nbr slices, writes out
        add(f"weighted_combine[{n_nbr}nbr]", ms, (n_nbr + 2) * numel * esz)
        # torch reference chain for the same math (what the reference runs,
        # mpi_ops.cc:119-155): n_nbr+1 fused-multiply passes over the data
        g2 = gathered.view(n_nbr, numel)

        def torch_chain():
            acc = self_t.mul(sw)
            for k in range(n_nbr):
                acc.add_(g2[k], alpha=w[k])
            return acc

        ms_t = time_kernel(torch_chain, iters)
        add(f"torch_chain[{n_nbr}nbr]", ms_t, (n_nbr + 2) * numel * esz)

    # fused average + SGD(momentum): reads p, 1 nbr slice, grad, mom; writes p, mom
    gathered = torch.randn(numel, device=dev).to(dtype)
    mom = torch.zeros_like(self_t)
    ms = time_kernel(
        lambda: _C.weighted_combine_sgd(
            self_t, 0.5, gathered, [0.5], grad, mom, 0.01, 0.9, 0.0, 0.0, False
        ),
        iters,
    )
    add("combine_sgd[1nbr+mom]", ms, 6 * numel * esz)

    if dtype in (torch.float32,):
        exp_avg = torch.zeros(numel, device=dev)
        exp_avg_sq = torch.zeros(numel, device=dev)
        ms = time_kernel(
            lambda: _C.weighted_combine_adam(
                self_t, 0.5, gathered, [0.5], grad, exp_avg, exp_avg_sq,
                1e-3, 0.9, 0.999, 1e-8, 0.0, 10,
            ),
            iters,
        )
        add("combine_adam[1nbr]", ms, 8 * numel * esz)

    if dtype in (torch.float32, torch.bfloat16):
        # fused residual add+ReLU (ResNet block join shape, channels_last)
        a4 = torch.randn(64, 256, 56, 56, device=dev).to(dtype).to(
            memory_format=torch.channels_last)
        b4 = torch.randn_like(a4)
        o4 = torch.empty_like(a4)
        g4 = torch.randn_like(a4)
        n4 = a4.numel()
        ms = time_kernel(lambda: _C.add_relu_fwd(o4, a4, b4), iters)
        add("add_relu_fwd[64x256x56x56]", ms, 3 * n4 * esz)
        ms = time_kernel(lambda: _C.relu_bwd_mask(o4, g4, a4), iters)
        add("relu_bwd_mask[64x256x56x56]", ms, 3 * n4 * esz)

    if dtype in (torch.float32, torch.bfloat16):
        # fused residual add + LayerNorm vs the torch chain, BERT-base
        # shape (bs16 x seq512 x H768)
        xb = torch.randn(16, 512, 768, device=dev, dtype=dtype)
        rb = torch.randn_like(xb)
        yb = torch.empty_like(xb)
        gln = torch.rand(768, device=dev) + 0.5
        bln = torch.randn(768, device=dev)
        nrows = 16 * 512
        mln = torch.empty(nrows, device=dev)
        sln = torch.empty(nrows, device=dev)
        nl = xb.numel()
        ms = time_kernel(lambda: _C.ln_add_fwd(yb, xb, rb, gln, bln, mln, sln, 1e-12), iters)
        add("ln_add_fwd[16x512x768]", ms, 3 * nl * esz)
        ms = time_kernel(
            lambda: torch.nn.functional.layer_norm(xb + rb, (768,), gln.to(dtype), bln.to(dtype)),
            iters,
        )
        add("torch_add_ln_fwd[16x512x768]", ms, 5 * nl * esz)
        dyb = torch.randn_like(xb)
        dxb = torch.empty_like(xb)
        dgln = torch.zeros(768, device=dev)
        dbln = torch.zeros(768, device=dev)
        scr = torch.empty(2 * 768 * _C.ln_bwd_scratch_rows(nrows), device=dev)
        ms = time_kernel(
            lambda: _C.ln_add_bwd(dxb, xb, rb, dyb, gln, mln, sln, dgln, dbln, scr),
            iters,
        )
        add("ln_add_bwd[16x512x768]", ms, 4 * nl * esz)

    ms = time_kernel(lambda: hip_ext.scale_put(out, self_t, 0.25), iters)
    add("scale_put", ms, 2 * numel * esz)
    ms = time_kernel(lambda: hip_ext.accum_put(out, self_t, 0.25), iters)
    add("accum_put", ms, 3 * numel * esz)
    ms = time_kernel(lambda: hip_ext.scale(out, 1.0001), iters)
    add("scale_inplace", ms, 2 * numel * esz)
    return rows


def main():
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--numel", type=int, default=25_557_032)  # ResNet50 params
    p.add_argument("--iters", type=int, default=50)
    p.add_argument("--json", action="store_true")
    args = p.parse_args()
    assert torch.cuda.is_available(), "needs an MI355X"
    dev = torch.device("cuda:0")
    rows = []
    for dtype in (torch.float32, torch.bfloat16, torch.float16, torch.float64):
        rows += run(args.numel, args.iters, dtype, dev)
    if args.json:
        print(json.dumps(rows))
    else:
        print(f"{'kernel':32s} {'dtype':9s} {'ms':>8s} {'GB/s':>9s}")
        for r in rows:
            print(f"{r['kernel']:32s} {r['dtype']:9s} {r['ms']:8.3f} {r['GBps']:9.0f}")


if __name__ == "__main__":
    main()
