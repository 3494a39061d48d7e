# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""BLUEFOG_CHECK_CONSISTENCY debug mode: ranks submitting ops in different
orders must fail fast with the diverging rank named, instead of
deadlocking (reference analog: coordinator validation,
operations.cc:293-433)."""

import pytest
import torch

from tests.util import run_dist


def w_consistent_ok():
    """Identical submission order on all ranks: checker stays silent."""
    import bluefog_amd as bf

    bf.init()
    bf.set_topology(bf.RingGraph(bf.size()))
    for i in range(10):
        x = torch.ones(4) * bf.rank()
        bf.neighbor_allreduce(x, name=f"t{i}")
    from bluefog_amd.ops.consistency import checker

    checker().flush()


def w_divergent_order_detected():
    """Rank 1 swaps two op names: every rank must raise a RuntimeError
    that names a diverging rank before any watchdog timeout."""
    import bluefog_amd as bf

    bf.init()
    bf.set_topology(bf.RingGraph(bf.size()))
    names = ["a", "b", "c", "d"]
    if bf.rank() == 1:
        names = ["a", "c", "b", "d"]  # divergence at op #1
    handles = []
    failed = None
    try:
        for n in names:
            x = torch.ones(4) * bf.rank()
            handles.append(bf.neighbor_allreduce_nonblocking(x, name=n))
        from bluefog_amd.ops.consistency import checker

        checker().flush()
    except RuntimeError as e:
        failed = str(e)
    assert failed is not None, "divergent op order was not detected"
    assert "rank 1" in failed or "diverge" in failed, failed


def w_shape_mismatch_detected():
    """Same op order but a different shape fingerprint on rank 0: the
    checker must catch it (recorded directly — actually posting
    mismatched p2p sizes would corrupt the comm layer under the test)."""
    import bluefog_amd as bf

    bf.init()
    from bluefog_amd.ops.consistency import checker

    shape = "(8,)" if bf.rank() != 0 else "(4,)"
    failed = None
    try:
        for i in range(3):
            checker().record(f"neighbor.allreduce.s{i}", 32,
                             f"shape={shape},dtype=torch.float32")
        checker().flush()
    except RuntimeError as e:
        failed = str(e)
    assert failed is not None, "shape divergence was not detected"
    # at ws=2 the tie goes to rank 0's order, so rank 1 is named diverging
    assert "rank 1" in failed and "shape=(4,)" in failed, failed


def test_consistency_ok():
    run_dist(w_consistent_ok, 2, env={"BLUEFOG_CHECK_CONSISTENCY": "4"},
             timeout=300)


def test_consistency_divergent_order():
    run_dist(w_divergent_order_detected, 2,
             env={"BLUEFOG_CHECK_CONSISTENCY": "8",
                  "BLUEFOG_CHECK_CONSISTENCY_TIMEOUT": "30"}, timeout=300)


def test_consistency_shape_mismatch():
    run_dist(w_shape_mismatch_detected, 2,
             env={"BLUEFOG_CHECK_CONSISTENCY": "8",
                  "BLUEFOG_CHECK_CONSISTENCY_TIMEOUT": "30"}, timeout=300)


def test_checker_disabled_by_default(monkeypatch):
    monkeypatch.delenv("BLUEFOG_CHECK_CONSISTENCY", raising=False)
    from bluefog_amd.ops import consistency

    consistency._reset_for_tests()
    assert not consistency.checker().enabled
    consistency._reset_for_tests()
