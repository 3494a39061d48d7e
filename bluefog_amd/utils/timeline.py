# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""Chrome-tracing timeline profiler.

Reference analog: bluefog/common/timeline.{h,cc} — same JSON trace format
(``chrome://tracing`` / perfetto loadable), one "pid" row per tensor name,
records drained by a writer thread so the hot path only pays a queue push.
Enabled by ``BLUEFOG_TIMELINE=<file>`` at init (the rank is appended to the
filename) or ``bf.timeline_start_activity`` / ``bf.timeline_context`` from
user code. GPU activities are stamped with host time after stream-ordered
posting; kernel-level GPU timing belongs to rocprofv3, which this format
complements rather than duplicates.
"""

import atexit
import json
import os
import queue
import threading
import time
from typing import Optional


def _roctx():
    """Optional roctx/NVTX bridge (SURVEY §5.1 bonus): BLUEFOG_ROCTX=1
    mirrors every activity as a roctx range so rocprofv3's marker trace
    (--marker-trace) aligns framework phases with kernel timelines.
    torch.cuda.nvtx lowers to roctx on ROCm builds."""
    if os.environ.get("BLUEFOG_ROCTX", "0") != "1":
        return None
    try:
        import torch

        return torch.cuda.nvtx
    except Exception:
        return None


class Timeline:
    def __init__(self):
        self._enabled = False
        self._file = None
        self._queue: "queue.Queue" = queue.Queue(maxsize=65536)
        self._writer: Optional[threading.Thread] = None
        self._pids = {}
        self._next_pid = 0
        self._t0 = time.monotonic_ns()
        self._lock = threading.Lock()
        self._active = {}
        self._gpu_anchors = {}  # device idx -> (anchor event, trace us)
        self._nvtx = _roctx()
        self._gpu_pending = []
        self._gpu_poller: Optional[threading.Thread] = None

    # ------------------------------------------------------------------
    def init(self, path: str, rank: int) -> None:
        if self._enabled:
            return
        fname = f"{path}_{rank}.json"
        self._file = open(fname, "w")
        self._file.write("[\n")
        self._enabled = True
        self._writer = threading.Thread(target=self._drain, daemon=True)
        self._writer.start()
        atexit.register(self.shutdown)

    @property
    def enabled(self) -> bool:
        return self._enabled

    def shutdown(self) -> None:
        if not self._enabled:
            return
        self._enabled = False
        self._queue.put(None)
        if self._writer is not None:
            self._writer.join(timeout=5.0)
        try:
            self._file.write(json.dumps({"name": "end", "ph": "i", "pid": 0, "ts": self._now_us()}))
            self._file.write("\n]\n")
            self._file.close()
        except Exception:
            pass

    # ------------------------------------------------------------------
    def _now_us(self) -> float:
        return (time.monotonic_ns() - self._t0) / 1000.0

    def _pid(self, tensor_name: str) -> int:
        with self._lock:
            pid = self._pids.get(tensor_name)
            if pid is None:
                pid = self._next_pid
                self._next_pid += 1
                self._pids[tensor_name] = pid
                self._emit(
                    {
                        "name": "process_name",
                        "ph": "M",
                        "pid": pid,
                        "args": {"name": tensor_name},
                    }
                )
            return pid

    def _emit(self, record: dict) -> None:
        if not self._enabled:
            return
        try:
            self._queue.put_nowait(record)
        except queue.Full:
            pass  # drop on overflow, like the reference's healthy-flag

    def _drain(self) -> None:
        while True:
            rec = self._queue.get()
            if rec is None:
                return
            try:
                self._file.write(json.dumps(rec) + ",\n")
            except Exception:
                return

    # ------------------------------------------------------------------
    def start_activity(self, tensor_name: str, activity: str) -> bool:
        if self._nvtx is not None:
            self._nvtx.range_push(f"{tensor_name}:{activity}")
        if not self._enabled:
            return False
        pid = self._pid(tensor_name)
        self._emit(
            {"name": activity, "ph": "B", "pid": pid, "tid": 0, "ts": self._now_us()}
        )
        self._active[tensor_name] = activity
        return True

    def end_activity(self, tensor_name: str) -> bool:
        if self._nvtx is not None:
            try:
                self._nvtx.range_pop()
            except Exception:
                pass
        if not self._enabled:
            return False
        if tensor_name not in self._active:
            return False
        pid = self._pid(tensor_name)
        self._emit({"ph": "E", "pid": pid, "tid": 0, "ts": self._now_us()})
        self._active.pop(tensor_name, None)
        return True

    def instant(self, tensor_name: str, activity: str) -> None:
        if not self._enabled:
            return
        pid = self._pid(tensor_name)
        self._emit(
            {"name": activity, "ph": "i", "pid": pid, "tid": 0, "ts": self._now_us(), "s": "t"}
        )

    # ------------------------------------------------------------------
    # GPU-completion timestamps (reference analog: pooled cudaEvents +
    # finalizer-thread timestamping, nccl_controller.cc:411-424,
    # 2066-2093). Each comm op records a hipEvent pair on the side stream;
    # a poller thread converts completed pairs to trace spans on tid=1
    # ("GPU" lane), so a trace shows *execution* windows — and therefore
    # backward/communication overlap — not host posting times.
    # ------------------------------------------------------------------
    def gpu_anchor(self, device) -> None:
        """Anchor GPU event time to the trace clock (once per device).
        Records + synchronizes one event — only called lazily on the first
        GPU span, never on the steady-state hot path."""
        import torch

        idx = device.index if device.index is not None else torch.cuda.current_device()
        if idx in self._gpu_anchors:
            return
        ev = torch.cuda.Event(enable_timing=True)
        ev.record(torch.cuda.current_stream(idx))
        ev.synchronize()
        self._gpu_anchors[idx] = (ev, self._now_us())

    def gpu_span(self, tensor_name: str, activity: str, device,
                 start_ev, end_ev) -> None:
        """Queue a recorded hipEvent pair; the poller emits the span when
        the end event retires (no host stall)."""
        if not self._enabled:
            return
        import torch

        idx = device.index if device.index is not None else torch.cuda.current_device()
        self.gpu_anchor(device)
        with self._lock:
            self._gpu_pending.append((tensor_name, activity, idx, start_ev, end_ev))
            if self._gpu_poller is None:
                self._gpu_poller = threading.Thread(
                    target=self._poll_gpu, daemon=True, name="bf-timeline-gpu"
                )
                self._gpu_poller.start()

    def _poll_gpu(self) -> None:
        while self._enabled:
            # 2 ms cadence while spans are in flight; back off 25x idle
            with self._lock:
                busy = bool(self._gpu_pending)
            time.sleep(0.002 if busy else 0.05)
            done = []
            with self._lock:
                still = []
                for item in self._gpu_pending:
                    try:
                        if item[4].query():
                            done.append(item)
                        else:
                            still.append(item)
                    except Exception:
                        pass  # drop spans whose events died
                self._gpu_pending = still
            for name, activity, idx, sev, eev in done:
                try:
                    anchor_ev, anchor_us = self._gpu_anchors[idx]
                    t0 = anchor_us + anchor_ev.elapsed_time(sev) * 1000.0
                    t1 = anchor_us + anchor_ev.elapsed_time(eev) * 1000.0
                    self._emit({
                        "name": activity, "ph": "X", "pid": self._pid(name),
                        "tid": 1, "ts": t0, "dur": max(t1 - t0, 0.0),
                        "args": {"lane": "gpu"},
                    })
                except Exception:
                    pass


_timeline = Timeline()


def timeline() -> Timeline:
    return _timeline


def maybe_init_from_env(rank: int) -> None:
    path = os.environ.get("BLUEFOG_TIMELINE", "")
    if path:
        _timeline.init(path, rank)
