# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""Hierarchical (machine-level) op tests with faked machines via
``BLUEFOG_NODES_PER_MACHINE`` (reference analog:
test/torch_hierarchical_test.py:39-46 + mpi_context.cc:320-337)."""

import pytest
import torch

from tests.util import run_dist

ENV2 = {"BLUEFOG_NODES_PER_MACHINE": "2"}


def w_machine_accessors():
    import bluefog_amd as bf

    bf.init()
    assert bf.local_size() == 2
    assert bf.machine_size() == bf.size() // 2
    assert bf.machine_rank() == bf.rank() // 2
    assert bf.local_rank() == bf.rank() % 2
    assert bf.is_homogeneous()


def w_hierarchical_local_allreduce():
    import bluefog_amd as bf

    bf.init()
    rank = bf.rank()
    t = torch.ones(6) * rank
    out = bf.allreduce(t, average=True, is_hierarchical_local=True)
    machine = rank // 2
    expected = (2 * machine + (2 * machine + 1)) / 2.0
    assert torch.allclose(out, torch.full((6,), expected)), (rank, out[0], expected)


def w_hierarchical_neighbor_allreduce():
    import bluefog_amd as bf

    bf.init()
    rank, size = bf.rank(), bf.size()
    nm = size // 2
    bf.set_machine_topology(bf.RingGraph(nm))
    t = torch.ones(4) * rank
    out = bf.hierarchical_neighbor_allreduce(t)
    machine = rank // 2

    def machine_sum(m):
        return 2 * m + (2 * m + 1)

    nbr_machines = bf.in_neighbor_machine_ranks()
    w = 1.0 / (len(nbr_machines) + 1)
    expected = (
        w * machine_sum(machine) + sum(w * machine_sum(m) for m in nbr_machines)
    ) / 2.0
    assert torch.allclose(out, torch.full((4,), expected), atol=1e-6), (
        rank,
        out[0].item(),
        expected,
    )


def w_hierarchical_dynamic():
    import bluefog_amd as bf
    import bluefog_amd.parallel.topology as tu

    bf.init()
    rank, size = bf.rank(), bf.size()
    local_size = bf.local_size()
    gen = tu.GetExp2DynamicSendRecvMachineRanks(
        size, local_size, rank, bf.local_rank()
    )
    machine = bf.machine_rank()

    def machine_sum(m):
        return sum(m * local_size + i for i in range(local_size))

    for _ in range(4):
        send_m, recv_m = next(gen)
        w = 1.0 / (len(recv_m) + 1)
        t = torch.ones(3, dtype=torch.float64) * rank
        out = bf.hierarchical_neighbor_allreduce(
            t,
            self_weight=w,
            src_machine_weights={m: w for m in recv_m},
            dst_machine_weights=send_m,
        )
        expected = (
            w * machine_sum(machine) + sum(w * machine_sum(m) for m in recv_m)
        ) / local_size
        assert torch.allclose(out, torch.full((3,), expected, dtype=torch.float64)), (
            rank,
            out[0].item(),
            expected,
        )


def w_hierarchical_optimizer():
    import torch.nn as nn
    import bluefog_amd as bf

    bf.init()
    bf.set_machine_topology(bf.RingGraph(bf.size() // 2))
    torch.manual_seed(4321)
    model = nn.Linear(16, 1, bias=False)
    g = torch.Generator().manual_seed(1234)
    w_star = torch.randn(16, 1, generator=g)
    g2 = torch.Generator().manual_seed(100 + bf.rank())
    A = torch.randn(128, 16, generator=g2)
    b = A @ w_star + 1e-3 * torch.randn(128, 1, generator=g2)
    opt = bf.DistributedAdaptWithCombineOptimizer(
        torch.optim.SGD(model.parameters(), lr=0.05),
        model=model,
        communication_type=bf.CommunicationType.hierarchical_neighbor_allreduce,
    )
    for _ in range(150):
        opt.zero_grad()
        loss = ((model(A) - b) ** 2).mean()
        loss.backward()
        opt.step()
    bf.allreduce_parameters(model.state_dict())
    with torch.no_grad():
        loss = ((model(A) - b) ** 2).mean().item()
    assert loss < 5e-3, loss


def test_machine_accessors():
    run_dist(w_machine_accessors, 4, env=ENV2)


def test_hierarchical_local_allreduce():
    run_dist(w_hierarchical_local_allreduce, 4, env=ENV2)


def test_hierarchical_neighbor_allreduce():
    run_dist(w_hierarchical_neighbor_allreduce, 4, env=ENV2)


def test_hierarchical_dynamic():
    run_dist(w_hierarchical_dynamic, 4, env=ENV2)


def test_hierarchical_optimizer():
    run_dist(w_hierarchical_optimizer, 4, env=ENV2, timeout=300)
