# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""Context/topology basics over real worlds (reference analog:
test/torch_basics_test.py)."""

import pytest
import torch

from tests.util import run_dist


def w_rank_size():
    import bluefog_amd as bf

    bf.init()
    assert 0 <= bf.rank() < bf.size()
    assert bf.local_size() >= 1
    assert not bf.mpi_threads_supported()  # no MPI here by design
    assert bf.nccl_built()


def w_set_load_topologies():
    import bluefog_amd as bf
    import bluefog_amd.parallel.topology as tu

    bf.init()
    size, rank = bf.size(), bf.rank()
    for builder in (
        tu.ExponentialTwoGraph,
        tu.RingGraph,
        tu.StarGraph,
        tu.FullyConnectedGraph,
        tu.MeshGrid2DGraph,
    ):
        topo = builder(size)
        assert bf.set_topology(topo)
        loaded = bf.load_topology()
        assert tu.IsTopologyEquivalent(topo, loaded)
        expected_in = sorted(r for r in topo.predecessors(rank) if r != rank)
        expected_out = sorted(r for r in topo.successors(rank) if r != rank)
        assert bf.in_neighbor_ranks() == expected_in
        assert bf.out_neighbor_ranks() == expected_out


def w_default_topology_is_exp2():
    import bluefog_amd as bf
    import bluefog_amd.parallel.topology as tu

    bf.init()
    assert tu.IsTopologyEquivalent(bf.load_topology(), tu.ExponentialGraph(bf.size()))


def w_weighted_topology():
    import bluefog_amd as bf
    import bluefog_amd.parallel.topology as tu

    bf.init()
    topo = tu.MeshGrid2DGraph(bf.size())
    bf.set_topology(topo, is_weighted=True)
    assert bf.is_topo_weighted()
    t = torch.ones(4, dtype=torch.float64) * (bf.rank() + 1)
    out = bf.neighbor_allreduce(t)
    self_w, nbr_w = tu.GetRecvWeights(topo, bf.rank())
    expected = self_w * (bf.rank() + 1) + sum(w * (r + 1) for r, w in nbr_w.items())
    assert torch.allclose(out, torch.full((4,), expected, dtype=torch.float64)), (
        bf.rank(),
        out,
        expected,
    )


def w_infer_round_trip():
    import bluefog_amd as bf

    bf.init()
    rank, size = bf.rank(), bf.size()
    # every rank sends to (rank+1)%size and (rank+2)%size
    dsts = sorted({(rank + 1) % size, (rank + 2) % size} - {rank})
    srcs = bf.InferSourceFromDestinationRanks(dsts)
    expected_srcs = sorted({(rank - 1) % size, (rank - 2) % size} - {rank})
    assert sorted(srcs) == expected_srcs, (srcs, expected_srcs)
    back = bf.InferDestinationFromSourceRanks(srcs)
    assert sorted(back) == dsts, (back, dsts)
    # with adjacency matrix
    srcs2, W = bf.InferSourceFromDestinationRanks(dsts, construct_adjacency_matrix=True)
    assert sorted(srcs2) == expected_srcs
    assert W.shape == (size, size)


def w_suspend_resume():
    import bluefog_amd as bf

    bf.init()
    bf.suspend()
    bf.resume()
    out = bf.allreduce(torch.ones(2))
    assert torch.allclose(out, torch.ones(2))


def w_skip_negotiate():
    import bluefog_amd as bf

    bf.init()
    assert bf.get_skip_negotiate_stage()  # default: program-order contract
    bf.set_skip_negotiate_stage(False)
    assert not bf.get_skip_negotiate_stage()
    bf.set_skip_negotiate_stage(True)


def test_rank_size():
    run_dist(w_rank_size, 2)


@pytest.mark.parametrize("ws", [2, 4])
def test_set_load_topologies(ws):
    run_dist(w_set_load_topologies, ws)


def test_default_topology_is_exp2():
    run_dist(w_default_topology_is_exp2, 4)


def test_weighted_topology():
    run_dist(w_weighted_topology, 4)


def test_infer_round_trip():
    run_dist(w_infer_round_trip, 4)


def test_suspend_resume():
    run_dist(w_suspend_resume, 2)


def test_skip_negotiate():
    run_dist(w_skip_negotiate, 2)


def test_single_process_init():
    # also valid without any env: world of one
    import subprocess, sys

    code = (
        "import torch, bluefog_amd as bf\n"
        "bf.init()\n"
        "assert bf.size() == 1 and bf.rank() == 0\n"
        "out = bf.neighbor_allreduce(torch.ones(3))\n"
        "assert torch.allclose(out, torch.ones(3))\n"
        "print('OK')\n"
    )
    r = subprocess.run(
        [sys.executable, "-c", code], capture_output=True, text=True, timeout=120
    )
    assert r.returncode == 0 and "OK" in r.stdout, r.stderr


def test_stall_watchdog_warns(monkeypatch, caplog):
    """Stalled-op detection (reference operations.cc:388-433): an op that
    never completes is reported with its name after the threshold."""
    import logging
    import time as _time

    from bluefog_amd.ops import handles as H

    monkeypatch.setattr(H, "_STALL_WARNING_TIME", 0.2)
    m = H.HandleManager()
    h = m.allocate("neighbor_allreduce.stalled_param")

    class _NeverDone:
        def is_completed(self):
            return False

    h.works = [_NeverDone()]
    from bluefog_amd.utils.logging import get_logger

    get_logger().addHandler(caplog.handler)  # logger is non-propagating
    with caplog.at_level(logging.WARNING, logger="bluefog_amd"):
        deadline = _time.time() + 5.0
        while _time.time() < deadline and not any(
            "stalled" in r.message for r in caplog.records
        ):
            _time.sleep(0.1)
    get_logger().removeHandler(caplog.handler)
    m.release(h.id)
    assert any(
        "neighbor_allreduce.stalled_param" in r.message and "stalled" in r.message
        for r in caplog.records
    )


def w_allreduce_parameters():
    import torch

    import bluefog_amd as bf

    bf.init()
    rank = bf.rank()
    params = {
        "w": torch.ones(4, 3) * (rank + 1.0),
        "b": torch.ones(2) * (10.0 * rank),
    }
    bf.allreduce_parameters(params)
    n = bf.size()
    expect_w = sum(r + 1.0 for r in range(n)) / n
    expect_b = sum(10.0 * r for r in range(n)) / n
    assert torch.allclose(params["w"], torch.full((4, 3), expect_w))
    assert torch.allclose(params["b"], torch.full((2,), expect_b))


def test_allreduce_parameters():
    from tests.util import run_dist

    run_dist(w_allreduce_parameters, 3)


def w_shutdown_clean():
    import torch

    import bluefog_amd as bf

    bf.init()
    t = torch.ones(8) * bf.rank()
    bf.allreduce(t, name="pre_shutdown")
    bf.win_create(t, "sw")
    bf.barrier()
    bf.shutdown()
    assert not bf._ctx().is_initialized()


def test_shutdown_clean():
    from tests.util import run_dist

    run_dist(w_shutdown_clean, 2)


def w_suspend_blocks_ops():
    """bf.suspend() must hold back new op submission until resume
    (reference: suspending the background comm thread)."""
    import threading
    import time as _time

    import torch

    import bluefog_amd as bf

    bf.init()
    bf.suspend()
    state = {"submitted": False}

    def submitter():
        t = torch.ones(4) * bf.rank()
        h = bf.allreduce_nonblocking(t, name="suspended_op")
        state["submitted"] = True
        state["out"] = bf.synchronize(h)

    th = threading.Thread(target=submitter)
    th.start()
    _time.sleep(0.3)
    assert not state["submitted"], "op went through while suspended"
    bf.resume()
    th.join(timeout=30)
    assert state["submitted"]
    n = bf.size()
    expected = sum(range(n)) / n
    assert torch.allclose(state["out"], torch.full((4,), expected))


def test_suspend_blocks_ops():
    from tests.util import run_dist

    run_dist(w_suspend_blocks_ops, 2)


def w_failed_op_releases_name():
    """A failing op must release its handle/name so a retry can reuse it."""
    import torch

    import bluefog_amd as bf
    from bluefog_amd.ops import engine
    from bluefog_amd.ops.handles import handle_manager

    bf.init()

    class _Boom:
        def is_completed(self):
            return True

        def wait(self):
            raise RuntimeError("transport exploded")

    h = handle_manager().allocate("retry_me")
    h.works = [_Boom()]
    try:
        engine.synchronize(h.id)
        raise AssertionError("expected the op failure to propagate")
    except RuntimeError:
        pass
    # name free again: a real op under the same name succeeds
    t = torch.ones(3) * bf.rank()
    out = bf.allreduce(t, name="retry_me")
    n = bf.size()
    assert torch.allclose(out, torch.full((3,), sum(range(n)) / n))


def test_failed_op_releases_name():
    from tests.util import run_dist

    run_dist(w_failed_op_releases_name, 2)


def test_reference_export_parity():
    """Every public symbol the reference's torch package exports must
    exist here (verified live against /root/reference when present)."""
    import os
    import re

    ref_init = "/root/reference/bluefog/torch/__init__.py"
    if not os.path.exists(ref_init):
        import pytest

        pytest.skip("reference tree not available on this host")
    src = open(ref_init).read()
    names = set()
    for m in re.finditer(r"from bluefog\.[\w.]+ import ([^#\n]+)", src):
        for part in m.group(1).split(","):
            part = part.strip().rstrip("\\").strip()
            if part and part.isidentifier():
                names.add(part)
    import bluefog_amd as bf

    missing = sorted(n for n in names if not hasattr(bf, n))
    assert not missing, f"missing reference exports: {missing}"
