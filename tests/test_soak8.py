# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""8-rank dress rehearsal of the flagship bench path on CPU/gloo.

The driver's 8-GPU scaling run executes exactly these combinations blind
(bench.py): AWC + dynamic one-peer exp2 with per-iteration weight updates,
and the win_put async-gossip rotation. This soak runs them at world size 8
for hundreds of steps on CPU, checks step-time stability, and logs the
per-op store round-trip counts that bound host overhead at scale
(reference bar: the >95% scaling headline, docs/performance.rst:37-46, is
won or lost on per-iteration host work)."""

import json
import os

import pytest
import torch
import torch.nn as nn

from tests.util import run_dist

STEPS = int(os.environ.get("BLUEFOG_SOAK_STEPS", "200"))


def _model():
    torch.manual_seed(77)
    return nn.Sequential(nn.Linear(32, 64), nn.ReLU(), nn.Linear(64, 32))


def _report(bf, label, times, extra=None):
    """Gather max step-time stats across ranks; rank 0 prints a JSON line
    (consumed by profiles/soak8_host_overhead.md)."""
    import torch.distributed as dist

    t = torch.tensor(times, dtype=torch.float64)
    stats = torch.tensor(
        [t.mean().item(), t.std().item(), t.max().item()], dtype=torch.float64
    )
    dist.all_reduce(stats, op=dist.ReduceOp.MAX)
    rpc = dict(bf._ctx().store.rpc_counts)
    total_rpc = sum(rpc.values())
    rec = {
        "soak": label,
        "world_size": bf.size(),
        "steps": len(times),
        "step_ms_mean_maxrank": stats[0].item() * 1e3,
        "step_ms_std_maxrank": stats[1].item() * 1e3,
        "step_ms_max_maxrank": stats[2].item() * 1e3,
        "store_rpcs_rank%d" % bf.rank(): total_rpc,
        "store_rpcs_per_step": total_rpc / max(len(times), 1),
        "rpc_breakdown": rpc,
    }
    if extra:
        rec.update(extra)
    if bf.rank() == 0:
        print("SOAK8 " + json.dumps(rec), flush=True)
    # stability: no step 50x slower than the mean (a stall would trip this)
    assert stats[2].item() < max(50 * stats[0].item(), 5.0), rec


def w_soak_awc_dynamic_exp2():
    """bench.py's default path: AWC fused-bucket submission + dynamic
    one-peer exp2 + per-iteration weight updates, 8 ranks."""
    import time

    import bluefog_amd as bf
    import bluefog_amd.parallel.topology as tu

    bf.init()
    topo = bf.ExponentialTwoGraph(bf.size())
    bf.set_topology(topo)
    model = _model()
    opt = bf.DistributedAdaptWithCombineOptimizer(
        torch.optim.SGD(model.parameters(), lr=0.01, momentum=0.9),
        model=model,
        communication_type=bf.CommunicationType.neighbor_allreduce,
    )
    import os as _os

    if _os.environ.get("BLUEFOG_FUSED_STEP") == "force":
        assert opt._fused == "sgd", "forced fused mode must engage on CPU"
    gen = tu.GetDynamicOnePeerSendRecvRanks(topo, bf.rank())
    bf.broadcast_parameters(model.state_dict(), root_rank=0)
    x = torch.randn(8, 32)
    times = []
    for _ in range(STEPS):
        send, recv = next(gen)
        w = 1.0 / (len(recv) + 1)
        opt.self_weight = w
        opt.src_weights = {r: w for r in recv}
        opt.dst_weights = send
        opt.enable_topo_check = False
        t0 = time.perf_counter()
        opt.zero_grad()
        (model(x) ** 2).mean().backward()
        opt.step()
        times.append(time.perf_counter() - t0)
    _report(bf, "awc_dynamic_exp2", times)


def w_soak_win_put_rotation():
    """bench.py's win_put path: async gossip with the destination rotated
    across the out-neighbor set every iteration (one xGMI link per step on
    GPU), 8 ranks — bounds the window control-plane host overhead."""
    import time

    import bluefog_amd as bf

    bf.init()
    bf.set_topology(bf.ExponentialTwoGraph(bf.size()))
    model = _model()
    opt = bf.DistributedWinPutOptimizer(
        torch.optim.SGD(model.parameters(), lr=0.01), model=model
    )
    bf.broadcast_parameters(model.state_dict(), root_rank=0)
    outs = bf.out_neighbor_ranks()
    x = torch.randn(8, 32)
    base_rpc = sum(bf._ctx().store.rpc_counts.values())
    times = []
    for i in range(STEPS):
        opt.dst_weights = {outs[i % len(outs)]: 1.0}
        t0 = time.perf_counter()
        opt.zero_grad()
        (model(x) ** 2).mean().backward()
        opt.step()
        times.append(time.perf_counter() - t0)
    win_rpc = sum(bf._ctx().store.rpc_counts.values()) - base_rpc
    _report(bf, "win_put_rotation", times,
            extra={"win_rpcs_per_step": win_rpc / STEPS})
    opt.unregister_window()


@pytest.mark.timeout(600)
def test_soak8_awc_dynamic_exp2():
    run_dist(w_soak_awc_dynamic_exp2, 8, timeout=540)


@pytest.mark.timeout(600)
def test_soak8_win_put_rotation():
    run_dist(w_soak_win_put_rotation, 8, timeout=540)


@pytest.mark.timeout(600)
def test_soak8_awc_with_consistency_checker():
    """The debug coordinator must stay silent on the legitimate dynamic
    exp2 schedule (its per-rank src/dst sets differ — they are detail,
    not digest) while adding bounded overhead."""
    run_dist(w_soak_awc_dynamic_exp2, 8,
             env={"BLUEFOG_CHECK_CONSISTENCY": "32",
                  "BLUEFOG_SOAK_STEPS": "60"}, timeout=540)


@pytest.mark.timeout(600)
def test_soak8_awc_fused_forced():
    """The EXACT 8-GPU flagship combination — fused bucket exchange +
    fused combine/step + dynamic one-peer exp2 — soaked at 8 ranks on
    CPU via the torch-op replica of the fused kernels
    (BLUEFOG_FUSED_STEP=force)."""
    run_dist(w_soak_awc_dynamic_exp2, 8,
             env={"BLUEFOG_FUSED_STEP": "force",
                  "BLUEFOG_SOAK_STEPS": "120"}, timeout=540)
