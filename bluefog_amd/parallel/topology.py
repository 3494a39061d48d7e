# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""Virtual-topology library: static graph families and dynamic one-peer
neighbor generators.

Reproduces the graph semantics of the reference's topology utilities
(reference: bluefog/common/topology_util.py:66-554) — same adjacency
matrices, same weights, same generator schedules — on top of our own
:class:`bluefog_amd.graph.DiGraph` (networkx is not required).

The flagship schedule for MI355X is :func:`GetDynamicOnePeerSendRecvRanks`
over :func:`ExponentialTwoGraph`: at 8 GPUs each rank cycles through its 3
Exp2 neighbors (distances 1, 2, 4), so each iteration's single send/recv
pair maps onto a distinct xGMI point-to-point link with zero contention —
the architectural reason neighbor averaging beats ring all-reduce on this
fabric (each of the 7 links carries the full payload in a ring).
"""

import math
from typing import Dict, Iterator, List, Optional, Tuple

import numpy as np

from bluefog_amd.graph import DiGraph, as_digraph

__all__ = [
    "IsTopologyEquivalent",
    "IsRegularGraph",
    "GetRecvWeights",
    "GetSendWeights",
    "ExponentialTwoGraph",
    "ExponentialGraph",
    "SymmetricExponentialGraph",
    "MeshGrid2DGraph",
    "StarGraph",
    "RingGraph",
    "FullyConnectedGraph",
    "GetDynamicOnePeerSendRecvRanks",
    "GetExp2DynamicSendRecvMachineRanks",
    "GetInnerOuterRingDynamicSendRecvRanks",
    "GetInnerOuterExpo2DynamicSendRecvRanks",
]


def IsTopologyEquivalent(topo1, topo2) -> bool:
    """True iff the two topologies have identical adjacency matrices (not
    isomorphism; reference topology_util.py:23-37)."""
    if topo1 is None or topo2 is None:
        return False
    g1, g2 = as_digraph(topo1), as_digraph(topo2)
    if g1.number_of_nodes() != g2.number_of_nodes():
        return False
    if g1.number_of_edges() != g2.number_of_edges():
        return False
    return bool((g1.to_numpy_array() == g2.to_numpy_array()).all())


def IsRegularGraph(topo) -> bool:
    """True iff every node has the same (in+out) degree."""
    g = as_digraph(topo)
    degree = g.degree(0)
    return all(g.degree(r) == degree for r in range(1, g.number_of_nodes()))


def GetRecvWeights(topo, rank: int) -> Tuple[float, Dict[int, float]]:
    """(self_weight, {src_rank: weight}) used on the receive side; weights
    come from column ``rank`` of the adjacency matrix."""
    g = as_digraph(topo)
    A = g.to_numpy_array()
    self_weight = 0.0
    neighbor_weights: Dict[int, float] = {}
    for src in g.predecessors(rank):
        if src == rank:
            self_weight = float(A[src, rank])
        else:
            neighbor_weights[src] = float(A[src, rank])
    return self_weight, neighbor_weights


def GetSendWeights(topo, rank: int) -> Tuple[float, Dict[int, float]]:
    """(self_weight, {dst_rank: weight}) used on the send side; weights come
    from row ``rank`` of the adjacency matrix."""
    g = as_digraph(topo)
    A = g.to_numpy_array()
    self_weight = 0.0
    neighbor_weights: Dict[int, float] = {}
    for dst in g.successors(rank):
        if dst == rank:
            self_weight = float(A[rank, dst])
        else:
            neighbor_weights[dst] = float(A[rank, dst])
    return self_weight, neighbor_weights


def _circulant(x: np.ndarray) -> DiGraph:
    """Graph whose row i is ``roll(x, i)`` — node i connects to (i+d) mod n
    for every nonzero x[d]."""
    size = len(x)
    topo = np.empty((size, size))
    for i in range(size):
        topo[i] = np.roll(x, i)
    return DiGraph(topo)


def isPowerOf(x, base: int) -> bool:
    assert isinstance(base, int), "Base has to be a integer."
    assert base > 1, "Base has to a interger larger than 1."
    assert x > 0
    return (base ** int(math.log(x, base))) == x


def ExponentialTwoGraph(size: int) -> DiGraph:
    """Each node i sends to (i + 2^k) mod size for all 2^k < size, uniform
    weights 1/(#neighbors+1) including the self loop."""
    assert size > 0
    x = np.array([1.0 if i & (i - 1) == 0 else 0.0 for i in range(size)])
    x /= x.sum()
    return _circulant(x)


def ExponentialGraph(size: int, base: int = 2) -> DiGraph:
    """Generalized exponential graph: distances that are powers of ``base``."""
    x = [1.0]
    for i in range(1, size):
        x.append(1.0 if isPowerOf(i, base) else 0.0)
    x = np.array(x)
    x /= x.sum()
    return _circulant(x)


def SymmetricExponentialGraph(size: int, base: int = 4) -> DiGraph:
    """Exponential distances mirrored around size/2."""
    x = [1.0]
    for i in range(1, size):
        index = i if i <= size // 2 else size - i
        x.append(1.0 if isPowerOf(index, base) else 0.0)
    x = np.array(x)
    x /= x.sum()
    return _circulant(x)


def MeshGrid2DGraph(size: int, shape: Optional[Tuple[int, int]] = None) -> DiGraph:
    """2D mesh with Metropolis–Hastings weights (policy 1 of
    arXiv:1702.05122); shape defaults to the two closest factors of size."""
    assert size > 0
    if shape is None:
        i = int(np.sqrt(size))
        while size % i != 0:
            i -= 1
        shape = (i, size // i)
    nrow, ncol = shape
    assert size == nrow * ncol, "The shape doesn't match the size provided."
    topo = np.zeros((size, size))
    for i in range(size):
        topo[i][i] = 1.0
        if (i + 1) % ncol != 0:
            topo[i][i + 1] = 1.0
            topo[i + 1][i] = 1.0
        if i + ncol < size:
            topo[i][i + ncol] = 1.0
            topo[i + ncol][i] = 1.0
    # Hastings rule: w(i,j) = 1/max(deg_i, deg_j) with self-including degree;
    # self weight absorbs the remainder so rows sum to 1.
    neighbors_with_self = [np.nonzero(topo[i])[0] for i in range(size)]
    for i in range(size):
        for j in neighbors_with_self[i]:
            if i != j:
                topo[i][j] = 1.0 / max(
                    len(neighbors_with_self[i]), len(neighbors_with_self[j])
                )
        topo[i][i] = 2.0 - topo[i].sum()
    return DiGraph(topo)


def StarGraph(size: int, center_rank: int = 0) -> DiGraph:
    """Bidirectional star around ``center_rank``."""
    assert size > 0
    topo = np.zeros((size, size))
    for i in range(size):
        topo[i, i] = 1 - 1 / size
        topo[center_rank, i] = 1 / size
        topo[i, center_rank] = 1 / size
    return DiGraph(topo)


def RingGraph(size: int, connect_style: int = 0) -> DiGraph:
    """Ring: style 0 = bidirectional, 1 = left only, 2 = right only."""
    assert size > 0
    assert 0 <= connect_style <= 2, (
        "connect_style has to be int between 0 and 2, where 0 for "
        "bi-connection, 1 for left connection, 2 for right connection."
    )
    if size == 1:
        return DiGraph(np.array([[1.0]]))
    if size == 2:
        return DiGraph(np.array([[0.5, 0.5], [0.5, 0.5]]))
    x = np.zeros(size)
    x[0] = 0.5
    if connect_style == 0:
        x[0] = 1 / 3.0
        x[-1] = 1 / 3.0
        x[1] = 1 / 3.0
    elif connect_style == 1:
        x[-1] = 0.5
    elif connect_style == 2:
        x[1] = 0.5
    return _circulant(x)


def FullyConnectedGraph(size: int) -> DiGraph:
    """Complete graph, uniform 1/size weights."""
    assert size > 0
    return _circulant(np.array([1 / size] * size))


# ---------------------------------------------------------------------------
# Dynamic one-peer generators
# ---------------------------------------------------------------------------


def GetDynamicOnePeerSendRecvRanks(
    topo, self_rank: int
) -> Iterator[Tuple[List[int], List[int]]]:
    """Cycle through the base topology's out-neighbors one at a time,
    clock-wise by circular distance; yields ([send_rank], recv_ranks) per
    iteration. Every rank running the same schedule keeps the send/recv sets
    globally consistent. (Reference semantics: topology_util.py:315-357.)"""
    g = as_digraph(topo)
    size = g.number_of_nodes()
    sorted_send_ranks = []
    for rank in range(size):
        sorted_ranks = sorted(
            g.successors(rank), key=lambda r, rk=rank: r - rk if r >= rk else r - rk + size
        )
        if sorted_ranks[0] == rank:
            sorted_ranks = sorted_ranks[1:]  # drop the self loop
        sorted_send_ranks.append(sorted_ranks)

    self_degree = g.out_degree(self_rank) - 1
    index = 0
    while True:
        send_rank = sorted_send_ranks[self_rank][index % self_degree]
        recv_ranks = []
        for other_rank in range(size):
            if other_rank == self_rank:
                continue
            degree = g.out_degree(other_rank) - 1
            if sorted_send_ranks[other_rank][index % degree] == self_rank:
                recv_ranks.append(other_rank)
        yield [send_rank], recv_ranks
        index += 1


def GetExp2DynamicSendRecvMachineRanks(
    world_size: int, local_size: int, self_rank: int, local_rank: int
) -> Iterator[Tuple[List[int], List[int]]]:
    """One-peer Exp2 schedule at machine granularity, for
    hierarchical_neighbor_allreduce (homogeneous placement only)."""
    assert (self_rank % local_size) == local_rank, (
        "world size must be a multiple of nodes_per_machine (homogeneous machines)."
    )
    assert (world_size % local_size) == 0, (
        "world size must be a multiple of nodes_per_machine (homogeneous machines)."
    )
    assert world_size > local_size, "It should be used under at least two machines case."

    machine_id = self_rank // local_size
    machine_size = world_size // local_size
    exp_2_size = int(np.log2(machine_size - 1)) if machine_size > 1 else 0
    index = 0
    while True:
        machine_dist = 2 ** (index % (exp_2_size + 1))
        send_machine_rank = (machine_id + machine_dist) % machine_size
        recv_machine_rank = (machine_id - machine_dist) % machine_size
        yield [send_machine_rank], [recv_machine_rank]
        index += 1


def GetInnerOuterRingDynamicSendRecvRanks(
    world_size: int, local_size: int, self_rank: int
) -> Iterator[Tuple[List[int], List[int]]]:
    """Inner-ring (within machine) / outer-ring (across machines) one-peer
    schedule: each iteration one designated local rank goes outside while the
    rest walk the inner ring skipping it."""
    num_machines = world_size // local_size
    nodes_per_machine = local_size
    assert world_size % local_size == 0, (
        "world size must be a multiple of nodes_per_machine (homogeneous machines)."
    )
    assert local_size > 2, (
        "nodes_per_machine must be at least 3 for the inner-outer schedule; "
        "with 2 or fewer use hierarchical_neighbor_allreduce or "
        "GetDynamicOnePeerSendRecvRanks instead."
    )

    index = 0
    while True:
        machine_id = self_rank // nodes_per_machine
        local_rank_id = self_rank % nodes_per_machine
        local_rank_to_go_outside_id = index % nodes_per_machine

        if local_rank_to_go_outside_id == local_rank_id:
            target_machine_id = (machine_id + 1) % num_machines
            send_rank = target_machine_id * nodes_per_machine + local_rank_id
            source_machine_id = (machine_id - 1) % num_machines
            recv_rank = source_machine_id * nodes_per_machine + local_rank_id
        else:
            target_local_rank_id = (local_rank_id + 1) % nodes_per_machine
            if target_local_rank_id == local_rank_to_go_outside_id:
                target_local_rank_id = (target_local_rank_id + 1) % nodes_per_machine
            send_rank = target_local_rank_id + machine_id * nodes_per_machine

            source_local_rank_id = (local_rank_id - 1) % nodes_per_machine
            if source_local_rank_id == local_rank_to_go_outside_id:
                source_local_rank_id = (source_local_rank_id - 1) % nodes_per_machine
            recv_rank = source_local_rank_id + machine_id * nodes_per_machine

        yield [send_rank], [recv_rank]
        index += 1


def GetInnerOuterExpo2DynamicSendRecvRanks(
    world_size: int, local_size: int, self_rank: int
) -> Iterator[Tuple[List[int], List[int]]]:
    """Inner-Exp2 / outer-Exp2 one-peer schedule (reference semantics,
    topology_util.py:466-554)."""
    num_machines = world_size // local_size
    nodes_per_machine = local_size
    assert world_size % local_size == 0, (
        "world size must be a multiple of nodes_per_machine (homogeneous machines)."
    )
    assert local_size > 2, (
        "nodes_per_machine must be at least 3 for the inner-outer schedule; "
        "with 2 or fewer use hierarchical_neighbor_allreduce or "
        "GetDynamicOnePeerSendRecvRanks instead."
    )

    exp_2_out_size = int(np.log2(num_machines - 1))
    if nodes_per_machine == 2:
        exp_2_in_size = 0
    else:
        # -2 because the rank going outside is excluded from the inner graph.
        exp_2_in_size = int(np.log2(nodes_per_machine - 2))

    index = 0
    while True:
        machine_id = self_rank // nodes_per_machine
        local_rank_id = self_rank % nodes_per_machine
        local_rank_to_go_outside_id = index % nodes_per_machine

        if local_rank_to_go_outside_id == local_rank_id:
            next_machine_dist = 2 ** (index % (exp_2_out_size + 1))
            target_machine_id = (machine_id + next_machine_dist) % num_machines
            send_rank = target_machine_id * nodes_per_machine + local_rank_id
            source_machine_id = (machine_id - next_machine_dist) % num_machines
            recv_rank = source_machine_id * nodes_per_machine + local_rank_id
        else:
            dist_to_out = (local_rank_to_go_outside_id - local_rank_id) % nodes_per_machine
            next_inner_dist = 2 ** (index % (exp_2_in_size + 1))
            if next_inner_dist >= dist_to_out:
                next_inner_dist += 1
            target_local_rank_id = (local_rank_id + next_inner_dist) % nodes_per_machine
            send_rank = target_local_rank_id + machine_id * nodes_per_machine

            reverse_inner_dist = 2 ** (index % (exp_2_in_size + 1))
            reverse_dist_to_out = (
                local_rank_id - local_rank_to_go_outside_id
            ) % nodes_per_machine
            if reverse_inner_dist >= reverse_dist_to_out:
                reverse_inner_dist += 1
            source_local_rank_id = (local_rank_id - reverse_inner_dist) % nodes_per_machine
            recv_rank = source_local_rank_id + machine_id * nodes_per_machine

        yield [send_rank], [recv_rank]
        index += 1
