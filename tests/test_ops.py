# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""Exact-value collective/neighbor op tests over real multi-process gloo
worlds (reference analog: test/torch_ops_test.py)."""

import numpy as np
import pytest
import torch

from tests.util import run_dist

DTYPES = [torch.float32, torch.float64]
HALF_DTYPES = [torch.float16, torch.bfloat16]


def _init_ring():
    import bluefog_amd as bf

    bf.init()
    bf.set_topology(bf.RingGraph(bf.size()))
    return bf


# --------------------------------------------------------------------------
# workers (module-level so spawn can import them)
# --------------------------------------------------------------------------


def w_allreduce():
    import bluefog_amd as bf

    bf.init()
    size, rank = bf.size(), bf.rank()
    for dtype in DTYPES + HALF_DTYPES:
        t = torch.ones(17, 3, dtype=dtype) * rank
        out = bf.allreduce(t, average=True)
        expected = sum(range(size)) / size
        assert torch.allclose(out, torch.full_like(t, expected), atol=1e-2), (
            dtype,
            out,
        )
        assert torch.equal(t, torch.ones(17, 3, dtype=dtype) * rank), "input modified"
    # in-place, sum
    t = torch.ones(5, dtype=torch.float32) * (rank + 1)
    bf.allreduce_(t, average=False)
    assert torch.allclose(t, torch.full((5,), float(sum(range(1, size + 1)))))


def w_broadcast():
    import bluefog_amd as bf

    bf.init()
    rank, size = bf.rank(), bf.size()
    for root in range(size):
        t = torch.arange(12, dtype=torch.float32).reshape(3, 4) * (rank + 1)
        out = bf.broadcast(t, root_rank=root)
        expected = torch.arange(12, dtype=torch.float32).reshape(3, 4) * (root + 1)
        assert torch.equal(out, expected)
        t2 = t.clone()
        bf.broadcast_(t2, root_rank=root)
        assert torch.equal(t2, expected)


def w_allgather():
    import bluefog_amd as bf

    bf.init()
    rank, size = bf.rank(), bf.size()
    t = torch.ones(2, 3) * rank
    out = bf.allgather(t)
    assert out.shape == (2 * size, 3)
    for r in range(size):
        assert torch.equal(out[2 * r : 2 * r + 2], torch.ones(2, 3) * r)
    # ragged first dims: rank r contributes r+1 rows
    t = torch.ones(rank + 1, 2) * rank
    out = bf.allgather(t)
    assert out.shape == (sum(r + 1 for r in range(size)), 2)
    off = 0
    for r in range(size):
        assert torch.equal(out[off : off + r + 1], torch.ones(r + 1, 2) * r)
        off += r + 1


def w_neighbor_allreduce_ring():
    bf = _init_ring()
    rank, size = bf.rank(), bf.size()
    for dtype in DTYPES:
        t = torch.ones(4, 2, dtype=dtype) * rank
        out = bf.neighbor_allreduce(t)
        left, right = (rank - 1) % size, (rank + 1) % size
        if size == 2:
            expected = (rank + left) / 2.0
        else:
            expected = (rank + left + right) / 3.0
        assert torch.allclose(out, torch.full_like(t, expected), atol=1e-6), (
            rank,
            out[0],
            expected,
        )
        assert out.shape == t.shape


def w_neighbor_allreduce_weighted():
    bf = _init_ring()
    rank, size = bf.rank(), bf.size()
    # explicit static weights
    left = (rank - 1) % size
    right = (rank + 1) % size
    if size == 2:
        src_weights = {left: 0.25}
    else:
        src_weights = {left: 0.25, right: 0.25}
    self_weight = 1.0 - sum(src_weights.values())
    t = torch.ones(3, dtype=torch.float64) * (rank + 1)
    out = bf.neighbor_allreduce(t, self_weight=self_weight, src_weights=src_weights)
    expected = self_weight * (rank + 1) + sum(w * (r + 1) for r, w in src_weights.items())
    assert torch.allclose(out, torch.full_like(t, expected)), (rank, out, expected)


def w_neighbor_allreduce_dynamic():
    import bluefog_amd as bf

    bf.init()
    rank, size = bf.rank(), bf.size()
    topo = bf.ExponentialTwoGraph(size)
    bf.set_topology(topo)
    import bluefog_amd.parallel.topology as tu

    gen = tu.GetDynamicOnePeerSendRecvRanks(topo, rank)
    for it in range(6):
        send_ranks, recv_ranks = next(gen)
        w = 1.0 / (len(recv_ranks) + 1)
        t = torch.ones(5, dtype=torch.float32) * rank
        out = bf.neighbor_allreduce(
            t,
            self_weight=w,
            src_weights={r: w for r in recv_ranks},
            dst_weights=send_ranks,
            enable_topo_check=True,
        )
        expected = w * rank + sum(w * r for r in recv_ranks)
        assert torch.allclose(out, torch.full_like(t, expected), atol=1e-6), (
            it,
            rank,
            out[0].item(),
            expected,
        )


def w_neighbor_allreduce_dst_weighting():
    import bluefog_amd as bf

    bf.init()
    rank, size = bf.rank(), bf.size()
    bf.set_topology(bf.FullyConnectedGraph(size))
    # every rank sends its value scaled by (dst+1)/10 to each dst
    dst_weights = {r: (r + 1) / 10.0 for r in range(size) if r != rank}
    src_weights = {r: 0.5 for r in range(size) if r != rank}
    t = torch.ones(4, dtype=torch.float64) * (rank + 1)
    out = bf.neighbor_allreduce(
        t, self_weight=0.5, src_weights=src_weights, dst_weights=dst_weights
    )
    expected = 0.5 * (rank + 1) + sum(
        0.5 * ((rank + 1) / 10.0) * (r + 1) for r in src_weights
    )
    assert torch.allclose(out, torch.full_like(t, expected)), (rank, out, expected)


def w_neighbor_allgather():
    bf = _init_ring()
    rank, size = bf.rank(), bf.size()
    t = torch.ones(2, 2) * rank
    out = bf.neighbor_allgather(t)
    in_nbrs = bf.in_neighbor_ranks()
    assert out.shape == (2 * len(in_nbrs), 2)
    for i, r in enumerate(in_nbrs):
        assert torch.equal(out[2 * i : 2 * i + 2], torch.ones(2, 2) * r)
    # ragged + dynamic: send to right, recv from left, rank r sends r+1 rows
    right, left = (rank + 1) % size, (rank - 1) % size
    t = torch.ones(rank + 1, 3) * rank
    out = bf.neighbor_allgather(t, src_ranks=[left], dst_ranks=[right])
    assert out.shape == (left + 1, 3)
    assert torch.equal(out, torch.ones(left + 1, 3) * left)


def w_pair_gossip():
    import bluefog_amd as bf

    bf.init()
    rank, size = bf.rank(), bf.size()
    assert size % 2 == 0
    partner = rank ^ 1
    t = torch.ones(6, dtype=torch.float32) * rank
    out = bf.pair_gossip(t, partner)
    assert torch.allclose(out, torch.full_like(t, (rank + partner) / 2.0))
    out2 = bf.pair_gossip(t, partner, self_weight=0.75, pair_weight=0.25)
    assert torch.allclose(out2, torch.full_like(t, 0.75 * rank + 0.25 * partner))


def w_nonblocking_many():
    """Many outstanding nonblocking ops at once (fusion-era stress)."""
    bf = _init_ring()
    rank, size = bf.rank(), bf.size()
    handles = []
    for i in range(50):
        t = torch.ones(11) * (rank + i)
        handles.append(bf.neighbor_allreduce_nonblocking(t, name=f"t{i}"))
    left, right = (rank - 1) % size, (rank + 1) % size
    nbrs = [left] if size == 2 else [left, right]
    for i, h in enumerate(handles):
        out = bf.synchronize(h)
        expected = (rank + i + sum(r + i for r in nbrs)) / (len(nbrs) + 1)
        assert torch.allclose(out, torch.full((11,), expected), atol=1e-5), (i, rank)


def w_poll_wait():
    bf = _init_ring()
    rank = bf.rank()
    t = torch.ones(3) * rank
    h = bf.neighbor_allreduce_nonblocking(t)
    # poll must eventually turn true, then synchronize returns the output
    import time

    for _ in range(10000):
        if bf.poll(h):
            break
        time.sleep(0.001)
    out = bf.synchronize(h)
    assert out.shape == t.shape


def w_barrier():
    import bluefog_amd as bf

    bf.init()
    bf.barrier()


def w_topo_check_mismatch():
    import bluefog_amd as bf

    bf.init()
    rank, size = bf.rank(), bf.size()
    # rank 0 claims it sends to 1, but 1 does not list 0 as a source
    if rank == 0:
        dst, src = [1], [1]
    else:
        dst, src = [0], []  # inconsistent on purpose
    try:
        bf.neighbor_allreduce(
            torch.ones(2),
            self_weight=0.5,
            src_weights={r: 0.5 for r in src},
            dst_weights=dst,
            enable_topo_check=True,
        )
        raise AssertionError("expected topo-check failure")
    except ValueError:
        pass




def w_consensus_convergence():
    """Iterated static-exp2 neighbor averaging drives every rank to the
    global mean — the framework's core semantic, end to end (reference
    examples/pytorch_average_consensus.py as a test)."""
    import bluefog_amd as bf

    bf.init()
    rank, size = bf.rank(), bf.size()
    bf.set_topology(bf.ExponentialTwoGraph(size))
    torch.manual_seed(100 + rank)
    x = torch.randn(64, dtype=torch.float64)
    mean = bf.allreduce(x, average=True, name="true_mean")
    for _ in range(40):
        x = bf.neighbor_allreduce(x)
    assert torch.allclose(x, mean, atol=1e-8), float((x - mean).abs().max())


def w_neighbor_allreduce_half():
    bf = _init_ring()
    rank, size = bf.rank(), bf.size()
    for dtype in HALF_DTYPES:
        t = torch.ones(33, dtype=dtype) * rank
        out = bf.neighbor_allreduce(t)
        left, right = (rank - 1) % size, (rank + 1) % size
        nbrs = [left] if size == 2 else [left, right]
        expected = (rank + sum(nbrs)) / (len(nbrs) + 1)
        assert out.dtype == dtype
        assert torch.allclose(
            out.float(), torch.full((33,), expected), atol=2e-2
        ), (dtype, rank, out[0].item(), expected)


def w_neighbor_allreduce_dims():
    bf = _init_ring()
    rank, size = bf.rank(), bf.size()
    left, right = (rank - 1) % size, (rank + 1) % size
    nbrs = [left] if size == 2 else [left, right]
    expected = (rank + sum(nbrs)) / (len(nbrs) + 1)
    for shape in [(1,), (23,), (4, 5), (2, 3, 4), (2, 2, 2, 2)]:
        t = torch.ones(shape, dtype=torch.float32) * rank
        out = bf.neighbor_allreduce(t)
        assert out.shape == t.shape
        assert torch.allclose(out, torch.full(shape, expected), atol=1e-6)


def w_inner_outer_dynamic():
    """Comm-exercise the InnerOuterRing/Expo2 dynamic generators (reference
    torch_ops_test dynamic move patterns): every iteration the claimed
    send/recv sets must be mutually consistent and produce exact averages."""
    import bluefog_amd as bf
    import bluefog_amd.parallel.topology as tu

    bf.init()
    rank, size = bf.rank(), bf.size()
    bf.set_topology(bf.ExponentialTwoGraph(size))
    for gen_fn in (
        tu.GetInnerOuterRingDynamicSendRecvRanks,
        tu.GetInnerOuterExpo2DynamicSendRecvRanks,
    ):
        gen = gen_fn(size, local_size=4, self_rank=rank)
        for it in range(6):
            send_ranks, recv_ranks = next(gen)
            w = 1.0 / (len(recv_ranks) + 1)
            t = torch.ones(7, dtype=torch.float64) * (rank + 1)
            out = bf.neighbor_allreduce(
                t,
                self_weight=w,
                src_weights={r: w for r in recv_ranks},
                dst_weights=send_ranks,
                enable_topo_check=True,
            )
            expected = w * (rank + 1) + sum(w * (r + 1) for r in recv_ranks)
            assert torch.allclose(out, torch.full_like(t, expected)), (
                gen_fn.__name__,
                it,
                rank,
            )


# --------------------------------------------------------------------------
# pytest entry points
# --------------------------------------------------------------------------


@pytest.mark.parametrize("ws", [2, 4])
def test_allreduce(ws):
    run_dist(w_allreduce, ws)


def test_broadcast():
    run_dist(w_broadcast, 2)


def test_allgather():
    run_dist(w_allgather, 3)


@pytest.mark.parametrize("ws", [2, 4])
def test_neighbor_allreduce_ring(ws):
    run_dist(w_neighbor_allreduce_ring, ws)


def test_neighbor_allreduce_weighted():
    run_dist(w_neighbor_allreduce_weighted, 4)


def test_neighbor_allreduce_dynamic():
    run_dist(w_neighbor_allreduce_dynamic, 4)


def test_neighbor_allreduce_dst_weighting():
    run_dist(w_neighbor_allreduce_dst_weighting, 3)


def test_neighbor_allgather():
    run_dist(w_neighbor_allgather, 4)


def test_pair_gossip():
    run_dist(w_pair_gossip, 2)


def test_nonblocking_many():
    run_dist(w_nonblocking_many, 4)


def test_poll_wait():
    run_dist(w_poll_wait, 2)


def test_barrier():
    run_dist(w_barrier, 4)


def test_topo_check_mismatch():
    run_dist(w_topo_check_mismatch, 2)

def test_consensus_convergence():
    run_dist(w_consensus_convergence, 4)


def test_neighbor_allreduce_half():
    run_dist(w_neighbor_allreduce_half, 2)


def test_neighbor_allreduce_dims():
    run_dist(w_neighbor_allreduce_dims, 2)


def test_inner_outer_dynamic():
    run_dist(w_inner_outer_dynamic, 8)


def w_weighted_topology_default_resolution():
    """set_topology(is_weighted=True): neighbor_allreduce with no explicit
    weights must use the graph's Metropolis-Hastings weights."""
    import bluefog_amd as bf
    import bluefog_amd.parallel.topology as tu

    bf.init()
    topo = tu.MeshGrid2DGraph(bf.size())
    bf.set_topology(topo, is_weighted=True)
    rank = bf.rank()
    t = torch.ones(6, dtype=torch.float64) * (rank + 1)
    out = bf.neighbor_allreduce(t)
    self_w, nbr_w = tu.GetRecvWeights(topo, rank)
    expected = self_w * (rank + 1) + sum(w * (r + 1) for r, w in nbr_w.items())
    assert torch.allclose(out, torch.full_like(t, expected)), (
        rank, float(out[0]), expected)


def w_error_paths():
    import bluefog_amd as bf

    bf.init()
    rank, size = bf.rank(), bf.size()
    t = torch.ones(3)
    # dst_weights containing self is rejected
    try:
        bf.neighbor_allreduce(
            t, self_weight=0.5, src_weights={(rank + 1) % size: 0.5},
            dst_weights=[rank],
        )
        raise AssertionError("self in dst_weights must be rejected")
    except ValueError:
        pass
    # src_weights must be a dict
    try:
        bf.neighbor_allreduce(t, self_weight=0.5, src_weights=[0], dst_weights=[0])
        raise AssertionError("non-dict src_weights must be rejected")
    except ValueError:
        pass
    # unknown window name
    try:
        bf.win_update("never_created")
        raise AssertionError("unknown window must raise")
    except ValueError:
        pass
    # duplicated window name
    assert bf.win_create(t, "dupwin")
    try:
        bf.win_create(t, "dupwin")
        raise AssertionError("duplicate window name must raise")
    except ValueError:
        pass
    bf.win_free("dupwin")
    bf.barrier()


def test_weighted_topology_default_resolution():
    run_dist(w_weighted_topology_default_resolution, 4)


def test_error_paths():
    run_dist(w_error_paths, 2)
