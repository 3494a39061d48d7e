# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""One-sided window op tests (reference analog: test/torch_win_ops_test.py):
lifecycle, update weights, put/get/accumulate exact values, versions, mutex,
associated-p push-sum consistency. CPU path (TCP window server); the GPU/IPC
path is covered in test_gpu_ops.py."""

import time

import numpy as np
import pytest
import torch

from tests.util import run_dist


def _init_ring():
    import bluefog_amd as bf

    bf.init()
    bf.set_topology(bf.RingGraph(bf.size()))
    return bf


def w_win_lifecycle():
    bf = _init_ring()
    t = torch.ones(4) * bf.rank()
    assert bf.win_create(t, "w0")
    assert bf.get_current_created_window_names() == ["w0"]
    if bf.size() > 2:  # for size 2 Star == Ring, set_topology short-circuits
        # topology change must be refused while a window exists
        assert not bf.set_topology(bf.StarGraph(bf.size()))
    assert bf.win_free("w0")
    assert bf.get_current_created_window_names() == []
    assert bf.set_topology(bf.StarGraph(bf.size()))


def w_win_update_default():
    bf = _init_ring()
    rank, size = bf.rank(), bf.size()
    t = torch.ones(6) * rank
    bf.win_create(t, "wu")
    # no puts yet: neighbor buffers hold creation-time clones of OUR tensor
    out = bf.win_update("wu")
    n = len(bf.in_neighbor_ranks())
    # buffers were initialized with self value => average == self value
    assert torch.allclose(out, torch.ones(6) * rank)
    bf.win_free("wu")


def w_win_put_update():
    bf = _init_ring()
    rank, size = bf.rank(), bf.size()
    t = torch.ones(5) * rank
    bf.win_create(t, "wp", zero_init=True)
    bf.barrier()
    assert bf.win_put(t, "wp")
    bf.barrier()
    out = bf.win_update("wp")
    # ring: in-neighbors put their rank values; buffers were zero before
    nbrs = bf.in_neighbor_ranks()
    w = 1.0 / (len(nbrs) + 1)
    expected = w * rank + sum(w * r for r in nbrs)
    assert torch.allclose(out, torch.full((5,), expected), atol=1e-6), (
        rank,
        out[0].item(),
        expected,
    )
    bf.barrier()
    bf.win_free("wp")


def w_win_put_weighted():
    bf = _init_ring()
    rank, size = bf.rank(), bf.size()
    t = torch.ones(3, dtype=torch.float64) * (rank + 1)
    bf.win_create(t, "wpw", zero_init=True)
    bf.barrier()
    dsts = {r: 0.5 for r in bf.out_neighbor_ranks()}
    bf.win_put(t, "wpw", dst_weights=dsts)
    bf.barrier()
    nbrs = bf.in_neighbor_ranks()
    nbr_w = {r: 1.0 for r in nbrs}
    out = bf.win_update("wpw", 1.0, nbr_w)
    expected = (rank + 1) + sum(0.5 * (r + 1) for r in nbrs)
    assert torch.allclose(out, torch.full((3,), expected, dtype=torch.float64)), (
        rank,
        out,
        expected,
    )
    bf.barrier()
    bf.win_free("wpw")


def w_win_put_self_weight():
    bf = _init_ring()
    rank = bf.rank()
    t = torch.ones(4) * 8.0
    bf.win_create(t, "wsw", zero_init=True)
    bf.barrier()
    bf.win_put(t, "wsw", self_weight=0.5)
    bf.barrier()
    # after put, the local tensor is scaled in place by self_weight
    assert torch.allclose(t, torch.full((4,), 4.0))
    bf.barrier()
    bf.win_free("wsw")


def w_win_accumulate():
    bf = _init_ring()
    rank, size = bf.rank(), bf.size()
    t = torch.ones(4) * (rank + 1)
    bf.win_create(t, "wa", zero_init=True)
    bf.barrier()
    for _ in range(3):
        bf.win_accumulate(t, "wa")
    bf.barrier()
    out = bf.win_update("wa", 1.0, {r: 1.0 for r in bf.in_neighbor_ranks()})
    expected = (rank + 1) + 3 * sum(r + 1 for r in bf.in_neighbor_ranks())
    assert torch.allclose(out, torch.full((4,), float(expected))), (rank, out, expected)
    bf.barrier()
    bf.win_free("wa")


def w_win_get():
    bf = _init_ring()
    rank, size = bf.rank(), bf.size()
    t = torch.ones(4, dtype=torch.float64) * (rank + 1)
    bf.win_create(t, "wg", zero_init=True)
    bf.barrier()
    assert bf.win_get("wg")
    bf.barrier()
    out = bf.win_update("wg", 1.0, {r: 1.0 for r in bf.in_neighbor_ranks()})
    expected = (rank + 1) + sum(r + 1 for r in bf.in_neighbor_ranks())
    assert torch.allclose(out, torch.full((4,), float(expected), dtype=torch.float64)), (
        rank,
        out,
        expected,
    )
    bf.barrier()
    bf.win_free("wg")


def w_win_version():
    bf = _init_ring()
    rank, size = bf.rank(), bf.size()
    t = torch.ones(2) * rank
    bf.win_create(t, "wv", zero_init=True)
    bf.barrier()
    ver = bf.get_win_version("wv")
    assert all(v == 0 for v in ver.values()), ver
    bf.win_put(t, "wv")
    bf.win_put(t, "wv")
    bf.barrier()
    ver = bf.get_win_version("wv")
    assert all(v == 2 for v in ver.values()), ver
    bf.win_update("wv")
    ver = bf.get_win_version("wv")
    assert all(v == 0 for v in ver.values()), ver
    bf.barrier()
    bf.win_free("wv")


def w_win_version_get_path():
    """win_get also counts as an update of the local buffer (split "get"
    counter, single-writer = the owner): version must rise on get and
    clear on win_update, mixed freely with incoming puts."""
    bf = _init_ring()
    rank = bf.rank()
    t = torch.ones(2) * (rank + 1)
    bf.win_create(t, "wvg")
    bf.barrier()
    src = bf.in_neighbor_ranks()[0]
    bf.win_get("wvg", src_weights={src: 1.0})
    ver = bf.get_win_version("wvg")
    assert ver[src] == 1, ver
    bf.barrier()
    bf.win_put(t, "wvg")  # each out-neighbor's buffer for us bumps too
    bf.barrier()
    ver = bf.get_win_version("wvg")
    # src slot saw our get (1) plus src's put (1); other in-neighbors saw
    # only their put
    assert ver[src] == 2, ver
    assert all(v == 1 for r, v in ver.items() if r != src), ver
    bf.win_update("wvg")
    assert all(v == 0 for v in bf.get_win_version("wvg").values())
    bf.barrier()
    bf.win_free("wvg")


def w_win_mutex():
    bf = _init_ring()
    rank, size = bf.rank(), bf.size()
    t = torch.zeros(1)
    bf.win_create(t, "wm")
    bf.barrier()
    # serialize increments of a store counter under the self-rank mutex of
    # rank 0's window: all ranks contend on the same mutex
    from bluefog_amd.ops.context import ctx

    store = ctx().store
    with bf.win_mutex("wm", ranks=[0]):
        v = store.counter("test/mutex/check")
        time.sleep(0.02)
        store.reset_counter("test/mutex/check", v + 1)
    bf.barrier()
    assert store.counter("test/mutex/check") == size
    bf.barrier()
    bf.win_free("wm")


def w_win_lock():
    """win_lock excludes mutex-honoring writers from this rank's window:
    a put issued under the peer's lock must not land until release."""
    bf = _init_ring()
    rank = bf.rank()
    t = torch.zeros(2)
    bf.win_create(t, "wl")
    bf.barrier()
    from bluefog_amd.ops.context import ctx

    store = ctx().store
    if rank == 0:
        with bf.win_lock("wl"):
            store.set("test/wl/locked", b"1")
            # peer's mutex-honoring put must block while we hold the epoch
            time.sleep(0.3)
            assert bf.get_win_version("wl")[1] == 0, "put landed inside the epoch"
        store.wait(["test/wl/done"], timeout_s=60)
        assert bf.get_win_version("wl")[1] == 1
    elif rank == 1:
        store.wait(["test/wl/locked"], timeout_s=60)
        bf.win_put(torch.ones(2), "wl", dst_weights={0: 1.0}, require_mutex=True)
        store.set("test/wl/done", b"1")
    bf.barrier()
    bf.win_free("wl")


def w_associated_p():
    bf = _init_ring()
    rank, size = bf.rank(), bf.size()
    bf.turn_on_win_ops_with_associated_p()
    try:
        t = torch.ones(3) * rank
        bf.win_create(t, "wap", zero_init=True)
        assert bf.win_associated_p("wap") == pytest.approx(1.0)
        bf.barrier()
        outdeg = len(bf.out_neighbor_ranks())
        w = 1.0 / (outdeg + 1)
        bf.win_accumulate(t, "wap", self_weight=w, dst_weights={r: w for r in bf.out_neighbor_ranks()})
        bf.barrier()
        bf.win_update_then_collect("wap")
        # mass conservation: sum of p over ranks stays == size
        p = torch.tensor([bf.win_associated_p("wap")])
        total = bf.allreduce(p, average=False)
        assert total.item() == pytest.approx(size, rel=1e-5), total
        bf.barrier()
        bf.win_free("wap")
    finally:
        bf.turn_off_win_ops_with_associated_p()


def w_push_sum_consistency():
    """Randomized push-sum over the extended-scalar pattern: after enough
    rounds every rank's corrected value approaches the global average
    (reference analog: torch_win_ops_test.py:780-863)."""
    import bluefog_amd as bf

    bf.init()
    size, rank = bf.size(), bf.rank()
    bf.set_topology(bf.ExponentialTwoGraph(size))
    x = torch.tensor([float(rank)], dtype=torch.float64)
    ext = torch.cat([x, torch.ones(1, dtype=torch.float64)])
    bf.win_create(ext, "ps", zero_init=True)
    bf.barrier()
    outdeg = len(bf.out_neighbor_ranks())
    w = 1.0 / (outdeg + 1)
    for _ in range(40):
        bf.win_accumulate(
            ext, "ps", dst_weights={r: w for r in bf.out_neighbor_ranks()},
            require_mutex=True,
        )
        bf.barrier()
        ext.mul_(w)
        ext = bf.win_update_then_collect("ps")
        bf.barrier()
    avg = ext[0] / ext[1]
    expected = sum(range(size)) / size
    assert abs(avg.item() - expected) < 1e-6, (rank, avg.item(), expected)
    bf.win_free("ps")


@pytest.mark.parametrize("ws", [2, 4])
def test_win_lifecycle(ws):
    run_dist(w_win_lifecycle, ws)


def test_win_update_default():
    run_dist(w_win_update_default, 4)


@pytest.mark.parametrize("ws", [2, 4])
def test_win_put_update(ws):
    run_dist(w_win_put_update, ws)


def test_win_put_weighted():
    run_dist(w_win_put_weighted, 4)


def test_win_put_self_weight():
    run_dist(w_win_put_self_weight, 2)


def test_win_accumulate():
    run_dist(w_win_accumulate, 4)


def test_win_get():
    run_dist(w_win_get, 4)


def test_win_version():
    run_dist(w_win_version, 2)


def test_win_version_get_path():
    run_dist(w_win_version_get_path, 4)


def test_win_mutex():
    run_dist(w_win_mutex, 4)


def test_win_lock():
    run_dist(w_win_lock, 2, timeout=300)


def test_associated_p():
    run_dist(w_associated_p, 4)


def test_push_sum_consistency():
    run_dist(w_push_sum_consistency, 4, timeout=300)


def w_win_multiwindow_stress():
    """Multi-window, multi-writer: 4 ranks, 3 windows each; every rank
    accumulates into BOTH its ring neighbors' windows while every rank
    concurrently collects its own — per (window, origin) totals must be
    exact at the end (no lost updates, no torn reads across windows
    sharing the single worker thread and the batched control plane)."""
    import time

    import bluefog_amd as bf

    bf.init()
    rank, size = bf.rank(), bf.size()
    bf.set_topology(bf.RingGraph(size))
    n_puts = 15
    names = [f"mw{w}" for w in range(3)]
    for name in names:
        bf.win_create(torch.zeros(64), name, zero_init=True)
    bf.barrier()
    dsts = {r: 1.0 for r in bf.out_neighbor_ranks()}
    ones = torch.ones(64)
    for i in range(n_puts):
        for name in names:
            bf.win_accumulate(ones, name, dst_weights=dsts, require_mutex=True)
        if i % 5 == 4:
            for name in names:
                # interleaved collects fold delivered values into the
                # window tensor in place — racing the writers on purpose
                bf.win_update_then_collect(name)
    bf.barrier()  # all accumulates delivered (win_accumulate blocks)
    deadline = time.time() + 120
    expected = float(n_puts * len(bf.in_neighbor_ranks()))
    for name in names:
        out = bf.win_update_then_collect(name)
        while time.time() < deadline and float(out[0]) < expected:
            out = bf.win_update_then_collect(name)
            time.sleep(0.001)
        # a lost update undershoots forever; a torn/duplicated read
        # overshoots — only the exact total passes
        assert torch.allclose(out, torch.full((64,), expected)), (
            name, float(out.min()), float(out.max()), expected,
        )
    bf.barrier()
    bf.win_free()


def test_win_multiwindow_stress():
    run_dist(w_win_multiwindow_stress, 4, timeout=300)


def w_win_mutex_stress():
    """Atomicity stress: rank 0 fires many win_accumulates while rank 1
    concurrently runs win_update_then_collect (which resets the buffer).
    Under the distributed mutex no update may be lost: at the end rank 1's
    window tensor must hold exactly the sum of every accumulate."""
    import bluefog_amd as bf

    bf.init()
    rank, size = bf.rank(), bf.size()
    bf.set_topology(bf.RingGraph(size))
    n_puts = 40
    t = torch.zeros(257)
    bf.win_create(t, "stress", zero_init=True)
    bf.barrier()
    if rank == 0:
        ones = torch.ones(257)
        for _ in range(n_puts):
            bf.win_accumulate(ones, "stress", dst_weights={1: 1.0},
                              require_mutex=True)
    else:
        import time

        # collect concurrently with the accumulates; every collected value
        # folds into the window tensor, so it must converge to exactly
        # n_puts — a lost update (accumulate racing the reset) would
        # undershoot forever, a torn read would overshoot
        deadline = time.time() + 120
        out = t
        while time.time() < deadline and out[0] < n_puts:
            out = bf.win_update_then_collect("stress")
            time.sleep(0.001)
        assert torch.allclose(out, torch.full((257,), float(n_puts))), (
            float(out.min()),
            float(out.max()),
            n_puts,
        )
    bf.barrier()
    bf.win_free()


def test_win_mutex_stress():
    run_dist(w_win_mutex_stress, 2, timeout=180.0)


def test_win_latency_probe_script():
    """The dev probe (scripts/win_latency_bench.py) self-spawns 2 ranks
    and prints a JSON report (reference analog: the standalone window
    experiments under scripts/)."""
    import json
    import os
    import subprocess
    import sys

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ)
    for k in ("RANK", "WORLD_SIZE", "MASTER_ADDR", "MASTER_PORT", "LOCAL_RANK"):
        env.pop(k, None)
    out = subprocess.run(
        [sys.executable, os.path.join(root, "scripts", "win_latency_bench.py"),
         "--sizes", "512", "--iters", "3"],
        capture_output=True, text=True, timeout=300, env=env, cwd=root,
    )
    assert out.returncode == 0, out.stderr[-1500:]
    # skip the "[Gloo] ..." banner lines; the report starts at a bare "["
    start = out.stdout.index("[\n")
    rows = json.loads(out.stdout[start:])
    assert rows and rows[0]["put_store_rpcs"] <= 2.0, rows
