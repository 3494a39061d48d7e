# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""Pure (no-process-group) tests of the topology library — adjacency/weight
semantics mirrored from the reference's graph families (reference test
analog: test/torch_basics_test.py topology portions)."""

import numpy as np
import pytest

import bluefog_amd.parallel.topology as tu
from bluefog_amd.graph import DiGraph


@pytest.mark.parametrize("size", [1, 2, 3, 4, 8, 12, 16])
def test_exponential_two_graph_row_stochastic(size):
    g = tu.ExponentialTwoGraph(size)
    A = g.to_numpy_array()
    assert A.shape == (size, size)
    np.testing.assert_allclose(A.sum(axis=1), np.ones(size))
    # successors of node i are (i + 2^k) mod size
    if size > 1:
        expected = {0} | {2**k for k in range(0, int(np.log2(size - 1)) + 1) if 2**k < size}
        assert set(g.successors(0)) == {e % size for e in expected}


def test_exponential_two_graph_8_neighbors():
    g = tu.ExponentialTwoGraph(8)
    # each rank: distances 1, 2, 4 plus self loop, uniform weight 1/4
    for i in range(8):
        assert sorted(g.successors(i)) == sorted({i, (i + 1) % 8, (i + 2) % 8, (i + 4) % 8})
    A = g.to_numpy_array()
    np.testing.assert_allclose(A[A > 0], 0.25)


@pytest.mark.parametrize(
    "builder",
    [
        lambda n: tu.ExponentialTwoGraph(n),
        lambda n: tu.ExponentialGraph(n, 2),
        lambda n: tu.SymmetricExponentialGraph(n, 4),
        lambda n: tu.MeshGrid2DGraph(n),
        lambda n: tu.StarGraph(n),
        lambda n: tu.RingGraph(n, 0),
        lambda n: tu.RingGraph(n, 1),
        lambda n: tu.RingGraph(n, 2),
        lambda n: tu.FullyConnectedGraph(n),
    ],
)
@pytest.mark.parametrize("size", [2, 4, 8, 12])
def test_all_families_row_stochastic(builder, size):
    A = builder(size).to_numpy_array()
    np.testing.assert_allclose(A.sum(axis=1), np.ones(size), atol=1e-12)


def test_ring_graph_values():
    g = tu.RingGraph(4, connect_style=0)
    A = g.to_numpy_array()
    np.testing.assert_allclose(A[0, [3, 0, 1]], [1 / 3, 1 / 3, 1 / 3])
    assert A[0, 2] == 0
    g1 = tu.RingGraph(4, connect_style=1)  # left
    assert set(g1.successors(0)) == {0, 3}
    g2 = tu.RingGraph(4, connect_style=2)  # right
    assert set(g2.successors(0)) == {0, 1}


def test_meshgrid_hastings_weights():
    g = tu.MeshGrid2DGraph(4, shape=(2, 2))
    A = g.to_numpy_array()
    # every interior weight 1/max(deg_i, deg_j); degrees all 3 (self + 2 nbrs)
    np.testing.assert_allclose(A.sum(axis=1), np.ones(4))
    assert A[0, 1] == pytest.approx(1 / 3)
    assert A[0, 0] == pytest.approx(1 / 3)


def test_star_graph():
    g = tu.StarGraph(5, center_rank=0)
    A = g.to_numpy_array()
    assert A[0, 0] == pytest.approx(1 / 5)
    for i in range(1, 5):
        assert A[i, 0] == pytest.approx(1 / 5)
        assert A[0, i] == pytest.approx(1 / 5)
        assert A[i, i] == pytest.approx(1 - 1 / 5)


def test_recv_send_weights():
    g = tu.RingGraph(4)
    self_w, nbr_w = tu.GetRecvWeights(g, 1)
    assert self_w == pytest.approx(1 / 3)
    assert nbr_w == {0: pytest.approx(1 / 3), 2: pytest.approx(1 / 3)}
    self_w2, send_w = tu.GetSendWeights(g, 1)
    assert self_w2 == pytest.approx(1 / 3)
    assert set(send_w) == {0, 2}


def test_topology_equivalence():
    assert tu.IsTopologyEquivalent(tu.RingGraph(4), tu.RingGraph(4))
    assert not tu.IsTopologyEquivalent(tu.RingGraph(4), tu.StarGraph(4))
    assert not tu.IsTopologyEquivalent(tu.RingGraph(4), tu.RingGraph(5))
    assert not tu.IsTopologyEquivalent(None, tu.RingGraph(4))


def test_is_regular():
    assert tu.IsRegularGraph(tu.RingGraph(6))
    assert tu.IsRegularGraph(tu.ExponentialTwoGraph(8))
    assert not tu.IsRegularGraph(tu.StarGraph(6))


@pytest.mark.parametrize("size", [4, 8, 11])
def test_dynamic_one_peer_consistency(size):
    """Every iteration: rank i sends to exactly one peer; the recv sets are
    the exact mirror across all ranks."""
    topo = tu.ExponentialTwoGraph(size)
    gens = [tu.GetDynamicOnePeerSendRecvRanks(topo, r) for r in range(size)]
    for _ in range(12):
        plan = [next(g) for g in gens]
        for me, (send, recv) in enumerate(plan):
            assert len(send) == 1
            # mirror check
            for s in send:
                assert me in plan[s][1]
            for r in recv:
                assert plan[r][0] == [me]


def test_dynamic_one_peer_exp2_cycle():
    topo = tu.ExponentialTwoGraph(8)
    gen = tu.GetDynamicOnePeerSendRecvRanks(topo, 0)
    sends = [next(gen)[0][0] for _ in range(6)]
    assert sends == [1, 2, 4, 1, 2, 4]


@pytest.mark.parametrize("ws,ls", [(8, 4), (12, 3), (16, 4)])
def test_inner_outer_ring_consistency(ws, ls):
    gens = [tu.GetInnerOuterRingDynamicSendRecvRanks(ws, ls, r) for r in range(ws)]
    for _ in range(10):
        plan = [next(g) for g in gens]
        for me, (send, recv) in enumerate(plan):
            assert len(send) == 1 and len(recv) == 1
            assert plan[send[0]][1] == [me]
            assert plan[recv[0]][0] == [me]


@pytest.mark.parametrize("ws,ls", [(16, 4), (32, 4)])
def test_inner_outer_expo2_consistency(ws, ls):
    gens = [tu.GetInnerOuterExpo2DynamicSendRecvRanks(ws, ls, r) for r in range(ws)]
    for _ in range(10):
        plan = [next(g) for g in gens]
        for me, (send, recv) in enumerate(plan):
            assert len(send) == 1 and len(recv) == 1
            assert plan[send[0]][1] == [me]
            assert plan[recv[0]][0] == [me]


def test_exp2_machine_ranks():
    gen = tu.GetExp2DynamicSendRecvMachineRanks(8, 2, self_rank=0, local_rank=0)
    out = [next(gen) for _ in range(4)]
    assert out[0] == ([1], [3])
    assert out[1] == ([2], [2])


def test_digraph_networkx_duck_typing():
    g = tu.RingGraph(4)
    g2 = DiGraph(g.to_numpy_array())
    assert tu.IsTopologyEquivalent(g, g2)
