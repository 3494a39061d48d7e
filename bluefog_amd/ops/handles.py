# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""Nonblocking-op handles.

Reference analog: bluefog/torch/handle_manager.{h,cc} plus the spin-wait in
mpi_ops.cc:549-555. Here a handle wraps stream-ordered work instead of a
background-thread status slot:

- GPU ops record a hipEvent on the side stream after the post-communication
  kernel; ``synchronize`` makes the *caller's* stream wait on that event
  (host never blocks), ``poll`` queries the event.
- CPU (gloo) ops keep their ``dist.Work`` list and run the post-op callback
  lazily on first synchronize/wait.
"""

import os
import threading
import time
from typing import Callable, List, Optional

import torch

from bluefog_amd.utils import metrics
from bluefog_amd.utils.logging import get_logger
from bluefog_amd.utils.timeline import timeline

# Stalled-op detection (reference analog: operations.cc:388-433 — the
# coordinator reports tensors stuck in negotiation for >60 s together with
# the ranks that never submitted them). Here a watchdog thread scans the
# outstanding-handle table; a handle alive past the threshold is reported
# once, with the op name, so a rank whose peers never posted the matching
# send/recv is diagnosable. BLUEFOG_STALL_WARNING_TIME seconds, 0 disables.
_STALL_WARNING_TIME = float(os.environ.get("BLUEFOG_STALL_WARNING_TIME", "60"))


class OpHandle:
    __slots__ = (
        "id",
        "name",
        "works",
        "finalize",
        "event",
        "future",
        "result",
        "done",
        "_lock",
    )

    def __init__(self, hid: int, name: str):
        self.id = hid
        self.name = name
        self.works: List = []
        self.finalize: Optional[Callable[[], torch.Tensor]] = None
        self.event: Optional[torch.cuda.Event] = None
        self.future = None  # concurrent.futures.Future (CPU window ops)
        self.result: Optional[torch.Tensor] = None
        self.done = False
        self._lock = threading.Lock()

    # -- completion --------------------------------------------------------
    def poll(self) -> bool:
        if self.done:
            return True
        if self.future is not None and not self.future.done():
            return False
        if self.event is not None:
            return bool(self.event.query())
        return all(w.is_completed() for w in self.works)

    def synchronize(self) -> torch.Tensor:
        with self._lock:
            if not self.done:
                if self.future is not None:
                    self.result = self.future.result()
                if self.event is not None:
                    # GPU path: stream-order the caller behind the post-op
                    # kernel; no host blocking.
                    torch.cuda.current_stream().wait_event(self.event)
                elif self.future is None:
                    for w in self.works:
                        w.wait()
                    if self.finalize is not None:
                        self.result = self.finalize()
                self.done = True
                timeline().end_activity(self.name)
        return self.result

    def wait_host(self) -> torch.Tensor:
        """Fully block the host until the op (incl. post-kernel) retired."""
        out = self.synchronize()
        if self.event is not None:
            self.event.synchronize()
        return out


class HandleManager:
    def __init__(self):
        self._lock = threading.Lock()
        self._next = 0
        self._handles = {}
        self._outstanding_names = set()
        self._birth = {}
        self._stall_reported = set()
        self._watchdog: Optional[threading.Thread] = None

    def _ensure_watchdog(self) -> None:
        # started lazily with the first handle; daemon thread, wakes every
        # few seconds — zero cost on the op hot path
        if self._watchdog is not None or _STALL_WARNING_TIME <= 0:
            return
        t = threading.Thread(target=self._watch, daemon=True, name="bf-stall-watchdog")
        self._watchdog = t
        t.start()

    def _watch(self) -> None:
        period = min(10.0, max(1.0, _STALL_WARNING_TIME / 6.0))
        while True:
            time.sleep(period)
            now = time.monotonic()
            with self._lock:
                stalled = [
                    (hid, h.name, now - self._birth[hid])
                    for hid, h in self._handles.items()
                    if now - self._birth.get(hid, now) > _STALL_WARNING_TIME
                    and hid not in self._stall_reported
                    and not h.done
                ]
                for hid, _, _ in stalled:
                    self._stall_reported.add(hid)
            for hid, name, age in stalled:
                if not self._handles.get(hid, OpHandle(-1, "")).poll():
                    metrics.record_stall(name)
                    get_logger().warning(
                        "op %r (handle %d) has not completed for %.0f s — "
                        "one or more peer ranks likely never submitted the "
                        "matching operation (stalled collective/p2p)%s",
                        name,
                        hid,
                        age,
                        self._peer_stall_report(name),
                    )

    def _peer_stall_report(self, stalled_name: str) -> str:
        """Best-effort: publish this rank's outstanding op names to the
        control store and read what peers have published, so the warning
        can name which ranks appear to be missing the stalled op
        (reference analog: the missing-rank stall report,
        operations.cc:388-433)."""
        try:
            from bluefog_amd.ops.context import ctx

            c = ctx()
            if not c.is_initialized() or c.size() == 1:
                return ""
            with self._lock:
                mine = sorted({h.name for h in self._handles.values() if not h.done})
            import json as _json

            c.store.set(f"stall/{c.rank()}", _json.dumps(mine).encode())
            peers = {}
            for r in range(c.size()):
                if r == c.rank():
                    continue
                if c.store.check([f"stall/{r}"]):
                    peers[r] = _json.loads(c.store.get(f"stall/{r}").decode())
            silent = [r for r in range(c.size())
                      if r != c.rank() and r not in peers]
            matching = [r for r, ops in peers.items() if stalled_name in ops]
            lines = [""]
            if matching:
                lines.append(
                    f"  peers also stalled on {stalled_name!r}: {matching} "
                    "(op posted everywhere; look for an earlier divergence)"
                )
            for r, ops in peers.items():
                if stalled_name not in ops:
                    lines.append(
                        f"  rank {r} does NOT have {stalled_name!r} outstanding "
                        f"(its oldest: {ops[:3]}) — likely the missing rank"
                    )
            if silent:
                lines.append(
                    f"  rank(s) {silent} report no stalls (running or dead)"
                )
            return "\n".join(lines)
        except Exception:  # never let diagnostics take down the watchdog
            return ""

    def allocate(self, name: str) -> OpHandle:
        with self._lock:
            if name in self._outstanding_names:
                raise ValueError(
                    f"Duplicated tensor name {name!r}: the same op name was "
                    "submitted again before the previous one finished "
                    "(reference DUPLICATE_NAME_ERROR, common.h:181-185)."
                )
            hid = self._next
            self._next += 1
            h = OpHandle(hid, name)
            self._handles[hid] = h
            self._outstanding_names.add(name)
            self._birth[hid] = time.monotonic()
            self._ensure_watchdog()
        return h

    def get(self, hid: int) -> OpHandle:
        with self._lock:
            h = self._handles.get(hid)
        if h is None:
            raise ValueError(f"Unknown bluefog handle {hid}")
        return h

    def release(self, hid: int) -> None:
        with self._lock:
            h = self._handles.pop(hid, None)
            self._birth.pop(hid, None)
            self._stall_reported.discard(hid)
            if h is not None:
                self._outstanding_names.discard(h.name)

    def poll(self, hid: int) -> bool:
        return self.get(hid).poll()

    def synchronize(self, hid: int) -> torch.Tensor:
        h = self.get(hid)
        try:
            out = h.synchronize()
        finally:
            # release even when the op failed, so the name can be reused
            # (a raised window/collective op would otherwise poison retries
            # with DUPLICATE_NAME_ERROR)
            if metrics.enabled():
                with self._lock:
                    birth = self._birth.get(hid)
                if birth is not None:
                    metrics.record_latency(h.name, time.monotonic() - birth)
            self.release(hid)
        return out


_handle_manager = HandleManager()


def handle_manager() -> HandleManager:
    return _handle_manager
