# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""Passive-target window transport for CPU tensors: a per-rank TCP server.

The reference implements CPU one-sided ops on MPI RMA windows
(mpi_controller.cc:952-1183). Without MPI, true passive-target semantics
need *some* agent at the target; here it is a tiny threaded TCP server per
process that applies put/accumulate/get against the window buffers under
the window's lock while the target's main thread trains undisturbed. The
GPU path does not use this — it writes peer HBM directly over xGMI (see
window_ipc.py); this server is the CPU-tensor and cross-node fallback.

Protocol: length-prefixed pickled dicts, one request per connection.
"""

import io
import os
import pickle
import socket
import socketserver
import struct
import threading


import torch


def _send_msg(sock: socket.socket, obj) -> None:
    payload = pickle.dumps(obj, protocol=pickle.HIGHEST_PROTOCOL)
    sock.sendall(struct.pack("<Q", len(payload)) + payload)


def _recv_msg(sock: socket.socket):
    hdr = _recv_exact(sock, 8)
    (n,) = struct.unpack("<Q", hdr)
    return pickle.loads(_recv_exact(sock, n))


def _recv_exact(sock: socket.socket, n: int) -> bytes:
    buf = io.BytesIO()
    got = 0
    while got < n:
        chunk = sock.recv(min(n - got, 1 << 20))
        if not chunk:
            raise ConnectionError("bluefog_amd window server: connection closed")
        buf.write(chunk)
        got += len(chunk)
    return buf.getvalue()


def _to_wire(t: torch.Tensor) -> torch.Tensor:
    return t.detach().cpu().contiguous()


class WindowServer:
    """Serves put/accumulate/get against this process's window buffers."""

    def __init__(self, registry):
        self._registry = registry
        outer = self

        class Handler(socketserver.BaseRequestHandler):
            def handle(self):
                try:
                    req = _recv_msg(self.request)
                    resp = outer._apply(req)
                    _send_msg(self.request, resp)
                except Exception as e:  # pragma: no cover - debug aid
                    try:
                        _send_msg(self.request, {"ok": False, "error": repr(e)})
                    except Exception:
                        pass

        class Server(socketserver.ThreadingTCPServer):
            daemon_threads = True
            allow_reuse_address = True

        # single-node by design; container hostnames often do not resolve
        # (use the loopback, matching the rendezvous contract). Bind to the
        # advertised host — the handler unpickles payloads, so listening on
        # all interfaces would expose a deserialization surface for nothing.
        self.host = os.environ.get("BLUEFOG_WIN_SERVER_HOST", "127.0.0.1")
        self._server = Server((self.host, 0), Handler)
        self.port = self._server.server_address[1]
        self._thread = threading.Thread(target=self._server.serve_forever, daemon=True)
        self._thread.start()

    def _apply(self, req: dict) -> dict:
        op = req["op"]
        win = self._registry.get(req["name"])
        with win.lock:
            if op == "put":
                buf = win.neighbor_buffer(req["origin"])
                buf.copy_(req["data"].to(buf.device))
                return {"ok": True}
            if op == "accum":
                buf = win.neighbor_buffer(req["origin"])
                buf.add_(req["data"].to(buf.device))
                return {"ok": True}
            if op == "get":
                return {"ok": True, "data": _to_wire(win.self_tensor)}
        return {"ok": False, "error": f"unknown op {op!r}"}

    def shutdown(self) -> None:
        try:
            self._server.shutdown()
            self._server.server_close()
        except Exception:  # pragma: no cover
            pass


class WindowClient:
    """Origin-side connection helper (one short-lived connection per op)."""

    def __init__(self, addr_by_rank):
        self._addrs = addr_by_rank  # {rank: (host, port)}

    def request(self, rank: int, payload: dict) -> dict:
        host, port = self._addrs[rank]
        with socket.create_connection((host, port), timeout=60.0) as sock:
            _send_msg(sock, payload)
            resp = _recv_msg(sock)
        if not resp.get("ok"):
            raise RuntimeError(
                f"bluefog_amd window op {payload.get('op')} on rank {rank} failed: "
                f"{resp.get('error')}"
            )
        return resp

    def put(self, rank: int, name: str, origin: int, data: torch.Tensor) -> None:
        self.request(rank, {"op": "put", "name": name, "origin": origin, "data": _to_wire(data)})

    def accum(self, rank: int, name: str, origin: int, data: torch.Tensor) -> None:
        self.request(rank, {"op": "accum", "name": name, "origin": origin, "data": _to_wire(data)})

    def get(self, rank: int, name: str) -> torch.Tensor:
        return self.request(rank, {"op": "get", "name": name})["data"]
