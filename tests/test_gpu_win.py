# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""GPU one-sided window tests: two processes share one MI355X (gloo control
backend — RCCL forbids two ranks per device) and exchange data through HIP
IPC peer buffers written by the native scale_put/accum_put kernels. This
validates the xGMI one-sided data plane end-to-end including cross-process
visibility of kernel stores into IPC-mapped memory."""

import pytest
import torch

from tests.util import run_dist

pytestmark = pytest.mark.gpu

ENV = {"BLUEFOG_BACKEND": "gloo"}


def w_gpu_win_put():
    import bluefog_amd as bf

    bf.init()
    rank, size = bf.rank(), bf.size()
    torch.cuda.set_device(0)
    bf.set_topology(bf.RingGraph(size))
    t = torch.ones(1 << 16, device="cuda") * (rank + 1)
    bf.win_create(t, "gwp", zero_init=True)
    from bluefog_amd.ops.window import registry

    win = registry().get("gwp")
    assert win.ipc is not None, "HIP IPC window transport must be active on GPU"
    bf.barrier()
    assert bf.win_put(t, "gwp")
    bf.barrier()
    out = bf.win_update("gwp", 1.0, {r: 1.0 for r in bf.in_neighbor_ranks()})
    expected = (rank + 1) + sum(r + 1 for r in bf.in_neighbor_ranks())
    torch.cuda.synchronize()
    assert torch.allclose(out, torch.full_like(out, float(expected))), (
        rank,
        out.flatten()[0].item(),
        expected,
    )
    bf.barrier()
    bf.win_free("gwp")


def w_gpu_win_accumulate_get():
    import bluefog_amd as bf

    bf.init()
    rank, size = bf.rank(), bf.size()
    torch.cuda.set_device(0)
    bf.set_topology(bf.RingGraph(size))
    t = torch.ones(4096, device="cuda", dtype=torch.float32) * (rank + 1)
    bf.win_create(t, "gwa", zero_init=True)
    bf.barrier()
    for _ in range(2):
        bf.win_accumulate(t, "gwa", dst_weights={r: 0.5 for r in bf.out_neighbor_ranks()})
    bf.barrier()
    out = bf.win_update("gwa", 1.0, {r: 1.0 for r in bf.in_neighbor_ranks()})
    expected = (rank + 1) + sum(1.0 * (r + 1) for r in bf.in_neighbor_ranks())
    torch.cuda.synchronize()
    assert torch.allclose(out, torch.full_like(out, float(expected))), (
        rank,
        out.flatten()[0].item(),
        expected,
    )
    bf.barrier()
    # now exercise win_get: fetch the peer's current (updated) tensor
    bf.win_get("gwa")
    bf.barrier()
    bf.win_free("gwa")


def w_gpu_win_staleness():
    """Repeated put/update cycles must never read stale peer data (checks
    kernel-boundary visibility of xGMI stores under the version protocol)."""
    import bluefog_amd as bf

    bf.init()
    rank, size = bf.rank(), bf.size()
    torch.cuda.set_device(0)
    bf.set_topology(bf.RingGraph(size))
    t = torch.zeros(1 << 14, device="cuda")
    bf.win_create(t, "gws", zero_init=True)
    bf.barrier()
    for it in range(20):
        val = float(it * size + rank)
        t.fill_(val)
        bf.win_put(t, "gws")
        bf.barrier()
        nbrs = bf.in_neighbor_ranks()
        out = bf.win_update("gws", 0.0, {r: 1.0 / len(nbrs) for r in nbrs})
        expected = sum(it * size + r for r in nbrs) / len(nbrs)
        torch.cuda.synchronize()
        assert torch.allclose(out, torch.full_like(out, expected)), (
            it,
            rank,
            out.flatten()[0].item(),
            expected,
        )
        bf.barrier()
    bf.win_free("gws")


def test_gpu_win_put():
    run_dist(w_gpu_win_put, 2, env=ENV, timeout=300)


def test_gpu_win_accumulate_get():
    run_dist(w_gpu_win_accumulate_get, 2, env=ENV, timeout=300)


def test_gpu_win_staleness():
    run_dist(w_gpu_win_staleness, 2, env=ENV, timeout=300)
