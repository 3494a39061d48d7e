# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""Optional Prometheus metrics (beyond reference parity — SURVEY.md §5.5
notes the reference ships none).

Off by default and zero-cost when off. Enable with
``BLUEFOG_METRICS_PORT=<base>`` before ``bf.init()`` (each rank serves
``/metrics`` on ``base + rank``) or programmatically::

    import bluefog_amd as bf
    bf.init()
    bf.start_metrics_server(9090)        # this rank only

Exported series (all labelled ``op``):
- ``bluefog_ops_total``        — nonblocking ops submitted
- ``bluefog_bytes_total``      — payload bytes handed to the data plane
- ``bluefog_op_seconds``       — submit→synchronize host latency histogram
- ``bluefog_stalled_ops_total``— ops flagged by the stall watchdog
"""

import os
import threading
from typing import Optional

_enabled = False
_lock = threading.Lock()
_counters = {}


def enabled() -> bool:
    return _enabled


def start_server(port: int) -> None:
    """Start the per-rank /metrics endpoint and enable recording."""
    global _enabled
    from prometheus_client import start_http_server

    with _lock:
        if not _enabled:
            start_http_server(port)
            _init_series()
            _enabled = True


def maybe_start_from_env(rank: int) -> None:
    base = os.environ.get("BLUEFOG_METRICS_PORT")
    if base:
        start_server(int(base) + rank)


def _init_series() -> None:
    from prometheus_client import Counter, Histogram

    _counters["ops"] = Counter(
        "bluefog_ops_total", "nonblocking ops submitted", ["op"]
    )
    _counters["bytes"] = Counter(
        "bluefog_bytes_total", "payload bytes handed to the data plane", ["op"]
    )
    _counters["latency"] = Histogram(
        "bluefog_op_seconds", "submit-to-synchronize host latency", ["op"],
        buckets=(1e-4, 5e-4, 1e-3, 5e-3, 1e-2, 5e-2, 0.1, 0.5, 1.0, 5.0),
    )
    _counters["stalls"] = Counter(
        "bluefog_stalled_ops_total", "ops reported stalled by the watchdog", ["op"]
    )


_FAMILIES = (
    "hierarchical.neighbor.allreduce",
    "neighbor.allreduce",
    "neighbor.allgather",
    "pair.gossip",
    "allreduce",
    "allgather",
    "broadcast",
    "win.put",
    "win.get",
    "win.accumulate",
)


def _op_family(name: str) -> str:
    # handle names are "<op-family>.<tensor-name>[.<seq>]"
    for fam in _FAMILIES:
        if name.startswith(fam):
            return fam
    return name.split(".")[0]


def record_submit(name: str, nbytes: Optional[int] = None) -> None:
    if not _enabled:
        return
    fam = _op_family(name)
    _counters["ops"].labels(fam).inc()
    if nbytes:
        _counters["bytes"].labels(fam).inc(nbytes)


def record_latency(name: str, seconds: float) -> None:
    if not _enabled:
        return
    _counters["latency"].labels(_op_family(name)).observe(seconds)


def record_stall(name: str) -> None:
    if not _enabled:
        return
    _counters["stalls"].labels(_op_family(name)).inc()
