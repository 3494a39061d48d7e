# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""Multi-process test harness.

The reference runs its suite under ``mpirun -np 4 pytest`` (Makefile:14-60)
— every test is a real multi-process job. Here each test spawns its own
world of N processes over 127.0.0.1 gloo (CPU) or RCCL (GPU, under
``-m gpu``), runs a module-level function in every rank, and propagates any
rank's failure."""

import multiprocessing as mp
import os
import socket
import sys
import traceback
from typing import Callable, Dict, Optional

_REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def free_port() -> int:
    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _entry(rank, world_size, port, fn_module, fn_name, args, env, errq):
    try:
        os.environ["RANK"] = str(rank)
        os.environ["LOCAL_RANK"] = str(rank)
        os.environ["WORLD_SIZE"] = str(world_size)
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        os.environ.setdefault("GLOO_SOCKET_IFNAME", "lo")
        for k, v in (env or {}).items():
            os.environ[k] = v
        sys.path.insert(0, _REPO_ROOT)
        import importlib

        fn = getattr(importlib.import_module(fn_module), fn_name)
        fn(*args)
        errq.put((rank, None))
    except Exception:
        errq.put((rank, traceback.format_exc()))
        sys.exit(1)


def run_dist(
    fn: Callable,
    world_size: int = 2,
    args: tuple = (),
    env: Optional[Dict[str, str]] = None,
    timeout: float = 180.0,
) -> None:
    """Run ``fn()`` (a module-level function) in ``world_size`` fresh
    processes with a gloo/RCCL rendezvous on 127.0.0.1. Rendezvous-level
    failures (a just-freed port grabbed by another process before the
    store binds it) get ONE retry on a fresh port; genuine test failures
    do not."""
    try:
        return _run_dist_once(fn, world_size, args, env, timeout)
    except (TimeoutError, AssertionError) as e:
        msg = str(e)
        transient = any(
            pat in msg
            for pat in (
                "Address already in use",
                "EADDRINUSE",
                "Connection refused",
                "Connection reset",
                "timed out",
            )
        )
        if not transient:
            raise
        return _run_dist_once(fn, world_size, args, env, timeout)


def _run_dist_once(
    fn: Callable,
    world_size: int,
    args: tuple,
    env: Optional[Dict[str, str]],
    timeout: float,
) -> None:
    ctx = mp.get_context("spawn")
    errq = ctx.Queue()
    port = free_port()
    procs = [
        ctx.Process(
            target=_entry,
            args=(r, world_size, port, fn.__module__, fn.__name__, args, env, errq),
        )
        for r in range(world_size)
    ]
    for p in procs:
        p.start()
    failures = []
    done = 0
    import queue as _q

    while done < world_size:
        try:
            rank, err = errq.get(timeout=timeout)
        except _q.Empty:
            for p in procs:
                p.terminate()
            raise TimeoutError(
                f"run_dist({fn.__name__}, world_size={world_size}) timed out"
            )  # noqa: TRY003 — retried once by run_dist for transient causes
        done += 1
        if err is not None:
            failures.append(f"--- rank {rank} ---\n{err}")
    for p in procs:
        p.join(timeout=30)
        if p.is_alive():
            p.terminate()
    if failures:
        raise AssertionError(
            f"{len(failures)} rank(s) failed in {fn.__name__}:\n" + "\n".join(failures)
        )
    if any(p.exitcode not in (0, None) for p in procs):
        codes = [p.exitcode for p in procs]
        raise AssertionError(f"nonzero exit codes in {fn.__name__}: {codes}")
