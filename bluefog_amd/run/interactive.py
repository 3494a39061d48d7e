# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""``ibfrun`` — interactive cluster for notebooks / REPLs.

Reference analog: bluefog/run/interactive_run.py, which starts an
ipyparallel controller + one engine per rank so users can drive BlueFog
ops interactively. ipyparallel is not part of this image, so the MI355X
build ships a self-contained equivalent:

- ``ibfrun start -np N [--daemonize]`` spawns N persistent worker
  processes wired with the same env rendezvous bfrun uses (so
  ``bf.init()`` works inside them) plus one TCP control socket per worker.
- From python/IPython, :class:`InteractiveClient` submits work::

      from bluefog_amd.run.interactive import InteractiveClient
      c = InteractiveClient()            # connects to the running cluster
      c.run(lambda: bf.init())           # on every rank
      outs = c.run(my_train_step, 32)    # returns [result_rank0, ...]
      c.run_code("x = bf.rank() * 2")    # exec statements; state persists
      c.pull("x")                        # -> [0, 2, 4, ...]

- ``ibfrun stop`` terminates the cluster (also reachable via
  ``InteractiveClient().shutdown()``).

Functions/args/results travel as pickles over localhost sockets with a
length prefix; each worker keeps a persistent namespace (``run_code`` /
``pull``) like an ipyparallel engine. Worker crashes surface as
ClusterError on the next call. A state file under ``~/.bluefog_amd``
advertises the ports, mirroring the reference's ipython-profile security
files (interactive_run.py:98-140).
"""

import argparse
import json
import os
import pickle
import signal
import socket
import struct
import subprocess
import sys
import time
from typing import Any, Callable, List, Optional

_STATE_DIR = os.path.expanduser(os.environ.get("BLUEFOG_IBFRUN_DIR", "~/.bluefog_amd"))


def _state_path(profile: str) -> str:
    return os.path.join(_STATE_DIR, f"ibfrun_{profile}.json")


# ---------------------------------------------------------------------------
# wire helpers
# ---------------------------------------------------------------------------


def _send_msg(sock: socket.socket, obj: Any) -> None:
    payload = pickle.dumps(obj, protocol=pickle.HIGHEST_PROTOCOL)
    sock.sendall(struct.pack("!Q", len(payload)) + payload)


def _recv_exact(sock: socket.socket, n: int) -> bytes:
    buf = b""
    while len(buf) < n:
        chunk = sock.recv(n - len(buf))
        if not chunk:
            raise ConnectionError("ibfrun worker closed the connection")
        buf += chunk
    return buf


def _recv_msg(sock: socket.socket) -> Any:
    (n,) = struct.unpack("!Q", _recv_exact(sock, 8))
    return pickle.loads(_recv_exact(sock, n))


# ---------------------------------------------------------------------------
# worker process
# ---------------------------------------------------------------------------


def _worker_main(control_port: int) -> None:
    """Accept one client at a time; execute submitted callables/code in a
    persistent namespace. RANK/WORLD_SIZE/... come from the spawn env."""
    ns: dict = {}
    srv = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
    srv.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
    srv.bind(("127.0.0.1", control_port))
    srv.listen(1)
    while True:
        conn, _ = srv.accept()
        try:
            while True:
                req = _recv_msg(conn)
                kind = req["kind"]
                if kind == "shutdown":
                    _send_msg(conn, {"ok": True, "value": None})
                    conn.close()
                    srv.close()
                    return
                try:
                    if kind == "call":
                        fn = req["fn"]
                        value = fn(*req["args"], **req["kwargs"])
                    elif kind == "code":
                        exec(compile(req["code"], "<ibfrun>", "exec"), ns)  # noqa: S102
                        value = None
                    elif kind == "pull":
                        value = ns[req["name"]]
                    elif kind == "ping":
                        value = os.environ.get("RANK")
                    else:
                        raise ValueError(f"unknown request kind {kind!r}")
                    _send_msg(conn, {"ok": True, "value": value})
                except BaseException as e:  # noqa: BLE001 — forwarded to client
                    _send_msg(conn, {"ok": False, "error": repr(e)})
        except (ConnectionError, EOFError):
            continue  # client went away; wait for the next one


# ---------------------------------------------------------------------------
# client
# ---------------------------------------------------------------------------


class ClusterError(RuntimeError):
    pass


class InteractiveClient:
    """Connects to an ``ibfrun start``ed cluster and runs work on all ranks."""

    def __init__(self, profile: str = "bluefog"):
        path = _state_path(profile)
        if not os.path.exists(path):
            raise ClusterError(
                f"no interactive cluster found ({path}); run `ibfrun start -np N` first"
            )
        with open(path) as f:
            self._state = json.load(f)
        self.profile = profile
        self._socks: List[socket.socket] = []
        for port in self._state["control_ports"]:
            s = socket.create_connection(("127.0.0.1", port), timeout=120)
            self._socks.append(s)

    @property
    def num_workers(self) -> int:
        return len(self._socks)

    def _broadcast(self, req: dict) -> List[Any]:
        for s in self._socks:
            _send_msg(s, req)
        # drain every rank before raising so a failure on one rank does not
        # leave stale responses queued on the others
        resps = [_recv_msg(s) for s in self._socks]
        errors = [
            f"rank {rank} failed: {r['error']}"
            for rank, r in enumerate(resps)
            if not r["ok"]
        ]
        if errors:
            raise ClusterError("; ".join(errors))
        return [r["value"] for r in resps]

    def run(self, fn: Callable, *args, **kwargs) -> List[Any]:
        """Execute ``fn(*args, **kwargs)`` on every rank; list of results."""
        return self._broadcast({"kind": "call", "fn": fn, "args": args, "kwargs": kwargs})

    def run_code(self, code: str) -> None:
        """Exec statements in each worker's persistent namespace."""
        self._broadcast({"kind": "code", "code": code})

    def pull(self, name: str) -> List[Any]:
        """Fetch a variable from each worker's namespace."""
        return self._broadcast({"kind": "pull", "name": name})

    def ping(self) -> List[Any]:
        return self._broadcast({"kind": "ping"})

    def shutdown(self) -> None:
        try:
            self._broadcast({"kind": "shutdown"})
        except (ClusterError, ConnectionError):
            pass
        for s in self._socks:
            s.close()
        try:
            os.remove(_state_path(self.profile))
        except OSError:
            pass

    def close(self) -> None:
        for s in self._socks:
            s.close()


# ---------------------------------------------------------------------------
# CLI
# ---------------------------------------------------------------------------


def _free_port() -> int:
    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def start_cluster(np_: int, profile: str = "bluefog", extra_env=None) -> dict:
    os.makedirs(_STATE_DIR, exist_ok=True)
    path = _state_path(profile)
    if os.path.exists(path):
        raise ClusterError(
            f"cluster state {path} already exists — run `ibfrun stop` first"
        )
    master_port = _free_port()
    control_ports = [_free_port() for _ in range(np_)]
    pids = []
    for rank in range(np_):
        env = dict(os.environ)
        env.update(
            RANK=str(rank),
            LOCAL_RANK=str(rank),
            WORLD_SIZE=str(np_),
            LOCAL_WORLD_SIZE=str(np_),
            MASTER_ADDR="127.0.0.1",
            MASTER_PORT=str(master_port),
        )
        env.update(extra_env or {})
        # give each worker its own log file: inheriting the launcher's
        # stdout/stderr pipes would keep them open past `ibfrun start`,
        # hanging any caller that waits for EOF
        log = open(os.path.join(_STATE_DIR, f"ibfrun_{profile}_worker{rank}.log"), "ab")
        proc = subprocess.Popen(
            [
                sys.executable,
                "-c",
                "from bluefog_amd.run.interactive import _worker_main; "
                f"_worker_main({control_ports[rank]})",
            ],
            env=env,
            start_new_session=True,
            stdout=log,
            stderr=subprocess.STDOUT,
        )
        log.close()
        pids.append(proc.pid)
    state = {
        "np": np_,
        "pids": pids,
        "control_ports": control_ports,
        "master_port": master_port,
    }
    with open(path, "w") as f:
        json.dump(state, f)
    # wait until every worker accepts connections
    deadline = time.time() + float(os.environ.get("BLUEFOG_IBFRUN_STARTUP_TIMEOUT", "180"))
    for port in control_ports:
        while True:
            try:
                with socket.create_connection(("127.0.0.1", port), timeout=1):
                    break
            except OSError:
                if time.time() > deadline:
                    raise ClusterError("ibfrun workers did not come up before the startup timeout")
                time.sleep(0.1)
    return state


def stop_cluster(profile: str = "bluefog") -> None:
    path = _state_path(profile)
    if not os.path.exists(path):
        print("ibfrun: no running cluster")
        return
    try:
        InteractiveClient(profile).shutdown()
    except (ClusterError, ConnectionError, OSError):
        with open(path) as f:
            state = json.load(f)
        for pid in state["pids"]:
            try:
                os.kill(pid, signal.SIGTERM)
            except ProcessLookupError:
                pass
        os.remove(path)


def main(argv: Optional[List[str]] = None) -> int:
    p = argparse.ArgumentParser(prog="ibfrun", description="Bluefog interactive cluster")
    sub = p.add_subparsers(dest="action", required=True)
    ps = sub.add_parser("start", help="start the interactive cluster")
    ps.add_argument("-np", "--num-proc", dest="np", type=int, required=True)
    ps.add_argument("--ipython-profile", dest="profile", default="bluefog")
    ps.add_argument("--extra-env", action="append", default=[], help="KEY=VALUE")
    pst = sub.add_parser("stop", help="stop the interactive cluster")
    pst.add_argument("--ipython-profile", dest="profile", default="bluefog")
    args = p.parse_args(argv)
    if args.action == "start":
        extra = dict(kv.split("=", 1) for kv in args.extra_env)
        state = start_cluster(args.np, args.profile, extra)
        print(
            f"ibfrun: started {args.np} workers (pids {state['pids']}); "
            "connect with bluefog_amd.run.interactive.InteractiveClient()"
        )
    else:
        stop_cluster(args.profile)
        print("ibfrun: stopped")
    return 0


if __name__ == "__main__":
    sys.exit(main())
