# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""HIP IPC window transport: true one-sided RMA over xGMI.

Each rank exports (a) its window's contiguous per-in-neighbor buffer block
and (b) the registered tensor itself through CUDA/HIP IPC (dmabuf on this
driver; torch's ForkingPickler reduction carries the handle bytes through
the TCP store). Every out-neighbor opens the destination's block once at
win_create and keeps the mapping; afterwards

- ``put``:  one ``scale_put`` kernel stores tensor*w straight into the
  destination GPU's slot for this rank — the stores traverse the xGMI link,
  the destination does nothing (compare the reference's passive-recv thread
  + ncclSend/Recv pair-communicator emulation, nccl_controller.cc:
  1261-1386, 1503-1887).
- ``accumulate``: read-modify-write on the peer slot (slot has exactly one
  writer — this rank — so plain loads/stores suffice).
- ``get``: reads the source's registered tensor over xGMI into the local
  buffer with the weight applied.

Kernel-boundary ordering: the worker thread launches each kernel on the
window stream and synchronizes it before bumping the version counter in the
TCP store, so a reader that observed the new version always sees completed
data (release on kernel retirement + host store acting as the flag carrier).
"""

import pickle
from typing import Dict

import torch

from bluefog_amd.ops import hip_ext

# importing torch.multiprocessing registers torch's ForkingPickler reducers
# (CUDA tensors serialize as IPC handles, not payload bytes)
import torch.multiprocessing  # noqa: F401
from multiprocessing.reduction import ForkingPickler


def ipc_export(t: torch.Tensor) -> bytes:
    return bytes(ForkingPickler.dumps(t))


def ipc_open(b: bytes) -> torch.Tensor:
    return pickle.loads(b)


class IpcWindowPeers:
    """Per-window peer mappings on this rank."""

    def __init__(self, win, c):
        self._win = win
        self._ctx = c
        me = c.rank()
        name = win.name
        store = c.store
        # export my block + self tensor
        store.set(f"win/{name}/ipc/{me}", ipc_export(win.block))
        store.set(f"win/{name}/ipcself/{me}", ipc_export(win.self_tensor))
        # open destinations' blocks (for put/accumulate) and sources' self
        # tensors (for get)
        need = [f"win/{name}/ipc/{d}" for d in win.out_ranks]
        need += [f"win/{name}/ipcself/{s}" for s in win.in_ranks]
        store.wait(need, timeout_s=120.0)

        topo = c.load_topology()
        self._peer_slot: Dict[int, torch.Tensor] = {}
        for dst in win.out_ranks:
            block = ipc_open(store.get(f"win/{name}/ipc/{dst}"))
            # slot layout contract: the destination built its block in
            # ITS ctx.in_neighbor_ranks() order, which set_topology defines
            # as sorted(predecessors) — recompute the same sorted list here
            # so our slot index matches the peer's layout exactly
            dst_in_ranks = sorted(r for r in topo.predecessors(dst) if r != dst)
            idx = dst_in_ranks.index(me)
            self._peer_slot[dst] = block[idx].view(win.self_tensor.shape)
        self._peer_self: Dict[int, torch.Tensor] = {
            src: ipc_open(store.get(f"win/{name}/ipcself/{src}"))
            for src in win.in_ranks
        }
        self._stream = torch.cuda.Stream()

    # -- data plane --------------------------------------------------------
    def put(self, dst: int, tensor: torch.Tensor, weight: float) -> None:
        with torch.cuda.stream(self._stream):
            hip_ext.scale_put(self._peer_slot[dst], tensor, weight)
        self._stream.synchronize()

    def accumulate(self, dst: int, tensor: torch.Tensor, weight: float) -> None:
        with torch.cuda.stream(self._stream):
            hip_ext.accum_put(self._peer_slot[dst], tensor, weight)
        self._stream.synchronize()

    def get(self, src: int, local_buf: torch.Tensor, weight: float) -> None:
        with torch.cuda.stream(self._stream):
            hip_ext.scale_put(local_buf, self._peer_self[src], weight)
        self._stream.synchronize()

    def close(self) -> None:
        self._peer_slot.clear()
        self._peer_self.clear()
