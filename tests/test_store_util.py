# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""Direct unit tests for the TCP-store control plane (ControlStore):
counters, multi-key round-trips, ticket-mutex fairness and the generation
barrier (reference analog: the MPI RMA spin-lock + control messages these
replace, mpi_controller.cc:1594-1663)."""

import threading

import torch.distributed as dist

from bluefog_amd.ops.store_util import ControlStore
from tests.util import free_port


def _mk_store(n=1):
    # one master store shared by n wrapper "ranks" (thread-level tests);
    # world_size=1 so the constructor does not wait for worker connections
    port = free_port()
    base = dist.TCPStore("127.0.0.1", port, 1, True)
    return [ControlStore(base, rank=r, size=n) for r in range(n)], base


def test_counters_and_kv():
    (s,), _base = _mk_store()
    assert s.counter("c") == 0  # add(0) creates
    assert s.add("c", 5) == 5
    assert s.counter("c") == 5
    s.reset_counter("c", 2)
    assert s.counter("c") == 2
    s.set("k", b"v")
    assert s.get("k") == b"v"
    assert s.check(["k"]) and not s.check(["absent"])


def test_multi_set_get_roundtrip():
    (s,), _base = _mk_store()
    pairs = {f"m/{i}": str(i).encode() for i in range(40)}
    s.multi_set(pairs)
    vals = s.multi_get(list(pairs))
    assert [int(v) for v in vals] == list(range(40))
    # counters written via multi_set stay add()-compatible
    s.multi_set({"mc": b"7"})
    assert s.add("mc", 3) == 10
    # empty calls are no-ops, not RPCs
    before = dict(s.rpc_counts)
    s.multi_set({})
    assert s.multi_get([]) == []
    assert dict(s.rpc_counts) == before


def test_mutex_mutual_exclusion_and_fairness():
    stores, _base = _mk_store(4)
    order = []
    lock_held = [False]

    def worker(i):
        st = stores[i]
        st.mutex_acquire("m")
        assert not lock_held[0], "two holders inside the critical section"
        lock_held[0] = True
        order.append(i)
        lock_held[0] = False
        st.mutex_release("m")

    threads = [threading.Thread(target=worker, args=(i,)) for i in range(4)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=60)
    assert sorted(order) == [0, 1, 2, 3]


def test_barrier_generations():
    stores, _base = _mk_store(3)
    results = []

    def worker(st):
        for gen in range(3):
            st.barrier("b", timeout_s=60)
            results.append(gen)

    threads = [threading.Thread(target=worker, args=(st,)) for st in stores]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=120)
    # every generation completed exactly size times
    assert sorted(results) == [0, 0, 0, 1, 1, 1, 2, 2, 2]


def test_mutex_timeout_does_not_wedge_the_queue():
    """A timed-out ticket is marked abandoned; later acquirers skip it
    instead of waiting forever behind a turn nobody will take."""
    (s,), _base = _mk_store()
    s.mutex_acquire("held")
    import pytest

    with pytest.raises(TimeoutError, match="abandoned"):
        s.mutex_acquire("held", timeout_s=0.2)
    s.mutex_release("held")
    # serving now points at the abandoned ticket; a fresh acquire must
    # skip past it and succeed
    s.mutex_acquire("held", timeout_s=5)
    s.mutex_release("held")
