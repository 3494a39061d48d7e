# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""TCP-store control plane.

The reference framework runs its whole control plane over MPI: negotiation
gathers/broadcasts, window-id sync, req/ack handshakes for the NCCL window
protocol, and a distributed mutex built from MPI_Fetch_and_op spins
(reference: bluefog/common/mpi_controller.cc:1594-1663, operations.cc:
853-1115). This framework has no MPI; every control-plane primitive is built
on one ``torch.distributed.TCPStore`` hosted by rank 0:

- atomic counters (``add``) -> distributed ticket mutex, window versions,
  push-sum bookkeeping, barrier generations;
- byte blobs (``set``/``get``) -> window registry metadata, IPC handles,
  CPU window data plane, debug coordinator messages.

Control traffic is a few hundred bytes per acquire/release — entirely off
the xGMI data plane.
"""


import time
from collections import Counter
from datetime import timedelta
from typing import List, Optional




class ControlStore:
    """Namespaced wrapper over a c10d Store with mutex/counter helpers."""

    #: seconds between polls while spinning on a ticket lock
    MUTEX_POLL_S = 0.0005

    def __init__(self, store, rank: int, size: int, prefix: str = "bf"):
        self._store = store
        self._rank = rank
        self._size = size
        self._prefix = prefix
        # round-trip accounting (one increment = one store RPC), for the
        # host-overhead soak tests and profiles
        self.rpc_counts = Counter()

    # -- raw kv ------------------------------------------------------------
    def _key(self, key: str) -> str:
        return f"{self._prefix}/{key}"

    def set(self, key: str, value: bytes) -> None:
        self.rpc_counts["set"] += 1
        self._store.set(self._key(key), value)

    def get(self, key: str) -> bytes:
        self.rpc_counts["get"] += 1
        return self._store.get(self._key(key))

    def wait(self, keys: List[str], timeout_s: Optional[float] = None) -> None:
        full = [self._key(k) for k in keys]
        if timeout_s is None:
            self._store.wait(full)
        else:
            self._store.wait(full, timedelta(seconds=timeout_s))

    def add(self, key: str, amount: int) -> int:
        self.rpc_counts["add"] += 1
        return self._store.add(self._key(key), amount)

    def multi_set(self, pairs: dict) -> None:
        """Write many keys in ONE store round-trip (values: bytes)."""
        if not pairs:
            return
        self.rpc_counts["multi_set"] += 1
        self._store.multi_set([self._key(k) for k in pairs], list(pairs.values()))

    def multi_get(self, keys: List[str]) -> List[bytes]:
        """Read many keys in ONE store round-trip. Every key must already
        exist (TCPStore blocks on missing keys)."""
        if not keys:
            return []
        self.rpc_counts["multi_get"] += 1
        return self._store.multi_get([self._key(k) for k in keys])

    def check(self, keys: List[str]) -> bool:
        self.rpc_counts["check"] += 1
        return self._store.check([self._key(k) for k in keys])

    def delete(self, key: str) -> bool:
        return self._store.delete_key(self._key(key))

    def counter(self, key: str) -> int:
        """Read an ``add``-maintained counter without changing it."""
        return self.add(key, 0)

    def reset_counter(self, key: str, value: int = 0) -> None:
        self._store.set(self._key(key), str(value))

    # -- distributed ticket mutex -----------------------------------------
    # Fair FIFO lock: acquire takes a ticket (atomic add) and spins until the
    # serving counter reaches it. Replaces the reference's MPI RMA spin lock.
    # A waiter that times out marks its ticket ABANDONED; when serving
    # reaches an abandoned ticket, exactly one live waiter advances past it
    # (guarded by a claim counter) — a timeout therefore never wedges the
    # queue for everyone behind it.
    def mutex_acquire(self, name: str, timeout_s: float = 60.0) -> None:
        ticket = self.add(f"mutex/{name}/next", 1) - 1
        deadline = time.monotonic() + timeout_s
        while True:
            serving = self.counter(f"mutex/{name}/serving")
            if serving == ticket:
                return
            if serving < ticket and self.check([f"mutex/{name}/abandon/{serving}"]):
                # skip an abandoned turn; the claim counter makes the
                # advance exactly-once even with many waiters
                if self.add(f"mutex/{name}/skipped/{serving}", 1) == 1:
                    self.add(f"mutex/{name}/serving", 1)
                continue
            if time.monotonic() > deadline:
                self.set(f"mutex/{name}/abandon/{ticket}", b"1")
                # our turn may have arrived in the same instant: do NOT
                # release it ourselves (a waiter could concurrently skip it
                # and double-advance); the skip path retires it safely
                raise TimeoutError(
                    f"bluefog_amd: timed out acquiring distributed mutex {name!r} "
                    f"(ticket {ticket}, serving {serving}); ticket abandoned"
                )
            time.sleep(self.MUTEX_POLL_S)

    def mutex_release(self, name: str) -> None:
        self.add(f"mutex/{name}/serving", 1)

    # -- generation barrier ------------------------------------------------
    def barrier(self, name: str, timeout_s: float = 300.0) -> None:
        """Store-side barrier independent of any process group (used during
        window lifecycle where no collective may be in flight)."""
        gen_key = f"barrier/{name}/gen"
        cnt_key = f"barrier/{name}/cnt"
        arrived = self.add(cnt_key, 1)
        gen_target, remainder = divmod(arrived - 1, self._size)
        if remainder == self._size - 1:
            self.add(gen_key, 1)
        deadline = time.monotonic() + timeout_s
        while self.counter(gen_key) < gen_target + 1:
            if time.monotonic() > deadline:
                raise TimeoutError(f"bluefog_amd: store barrier {name!r} timed out")
            time.sleep(self.MUTEX_POLL_S)
