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
        if not self._enabled:
            return False
        pid = self._pid(tensor_name)
        self._emit(
            {"name": activity, "ph": "B", "pid": pid, "tid": 0, "ts": self._now_us()}
        )
        self._active[tensor_name] = activity
        return True

    def end_activity(self, tensor_name: str) -> bool:
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


_timeline = Timeline()


def timeline() -> Timeline:
    return _timeline


def maybe_init_from_env(rank: int) -> None:
    path = os.environ.get("BLUEFOG_TIMELINE", "")
    if path:
        _timeline.init(path, rank)
