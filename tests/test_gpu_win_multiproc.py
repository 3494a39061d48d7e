# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""Multi-process GPU window tests on one device.

Two ranks share cuda:0 with the gloo control backend (BLUEFOG_BACKEND=gloo)
so the HIP-IPC window data plane — export via the TCP store, peer open,
``scale_put``/``accum_put`` kernels writing into the *other process's*
buffer — runs for real (window_ipc.py). On an 8-GPU node the same code maps
buffers across xGMI; same-device IPC exercises every line of the transport.
"""

import os
import socket
import subprocess
import sys

import pytest
import torch

pytestmark = pytest.mark.gpu

_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

_WORKER = r"""
import os, sys
sys.path.insert(0, os.environ["BF_ROOT"])
import torch
import bluefog_amd as bf

bf.init()
rank, size = bf.rank(), bf.size()
torch.cuda.set_device(0)
dev = torch.device("cuda:0")
bf.set_topology(bf.RingGraph(size))

# ---- win_put + win_update ------------------------------------------------
t = torch.ones(4096, device=dev) * (rank + 1.0)
assert bf.win_create(t, "w")
h = bf.win_put_nonblocking(t, "w")
bf.win_wait(h)
bf.barrier()
torch.cuda.synchronize()
# versions count un-acknowledged puts; the peer's put is pending
ver = bf.get_win_version("w")
assert all(v >= 1 for v in ver.values()), ver
out = bf.win_update("w")
# ring(2): one in-neighbor; uniform weights 1/2 each -> (1+2)/2 = 1.5
expected = ( (rank + 1.0) + (2.0 - rank) ) / 2.0
assert torch.allclose(out, torch.full_like(out, expected)), (
    rank, float(out.mean()))
# win_update acknowledged the put
ver = bf.get_win_version("w")
assert all(v == 0 for v in ver.values()), ver

# ---- win_accumulate ------------------------------------------------------
t2 = torch.ones(1000, device=dev) * (rank + 1.0)
assert bf.win_create(t2, "acc", zero_init=True)
h = bf.win_accumulate_nonblocking(t2, "acc")
bf.win_wait(h)
bf.barrier()
torch.cuda.synchronize()
# neighbor buffer now holds the peer's accumulated value; update with
# explicit weights: 0.5*self + 0.5*neighbor_buffer
out2 = bf.win_update("acc", self_weight=0.5,
                     neighbor_weights={1 - rank: 0.5})
expected2 = 0.5 * (rank + 1.0) + 0.5 * (2.0 - rank)
assert torch.allclose(out2, torch.full_like(out2, expected2)), (
    rank, float(out2.mean()))

# ---- win_get -------------------------------------------------------------
t3 = torch.ones(512, device=dev) * float(10 + rank)
assert bf.win_create(t3, "g")
bf.barrier()
h = bf.win_get_nonblocking("g")
bf.win_wait(h)
out3 = bf.win_update_then_collect("g")
bf.barrier()

bf.win_free()
print(f"MULTIPROC_WIN_OK rank={rank}", flush=True)
"""


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def test_gpu_ipc_windows_two_ranks_one_device():
    if not torch.cuda.is_available():
        pytest.skip("needs GPU")
    port = _free_port()
    procs = []
    for rank in range(2):
        env = dict(os.environ)
        env.update(
            BF_ROOT=_ROOT,
            RANK=str(rank),
            LOCAL_RANK=str(rank),
            WORLD_SIZE="2",
            MASTER_ADDR="127.0.0.1",
            MASTER_PORT=str(port),
            BLUEFOG_BACKEND="gloo",  # control plane; data plane = HIP IPC
            HSA_ENABLE_IPC_MODE_LEGACY="0",
        )
        procs.append(
            subprocess.Popen(
                [sys.executable, "-c", _WORKER],
                env=env,
                stdout=subprocess.PIPE,
                stderr=subprocess.STDOUT,
            )
        )
    outs = []
    for p in procs:
        try:
            out, _ = p.communicate(timeout=300)
        except subprocess.TimeoutExpired:
            p.kill()
            out, _ = p.communicate()
        outs.append(out.decode(errors="replace"))
    for rank, (p, out) in enumerate(zip(procs, outs)):
        assert p.returncode == 0, f"rank {rank} failed:\n{out}"
        assert f"MULTIPROC_WIN_OK rank={rank}" in out, out


_OPT_WORKER = r"""
import os, sys
sys.path.insert(0, os.environ["BF_ROOT"])
import torch
import torch.nn as nn
import bluefog_amd as bf

bf.init()
rank, size = bf.rank(), bf.size()
torch.cuda.set_device(0)
dev = torch.device("cuda:0")
bf.set_topology(bf.RingGraph(size))

torch.manual_seed(3)
model = nn.Sequential(nn.Linear(32, 64), nn.ReLU(), nn.Linear(64, 1)).to(dev)
opt = bf.DistributedWinPutOptimizer(
    torch.optim.SGD(model.parameters(), lr=0.03), model=model
)
bf.broadcast_parameters(model.state_dict(), root_rank=0)

# synthetic linear regression, different shard per rank
g = torch.Generator().manual_seed(50 + rank)
X = torch.randn(256, 32, generator=g).to(dev)
w_true = torch.arange(32, dtype=torch.float32).to(dev) / 32.0
Y = (X @ w_true).unsqueeze(1) + 0.01 * torch.randn(256, 1, generator=g).to(dev)

loss0 = None
for step in range(60):
    opt.zero_grad()
    loss = ((model(X) - Y) ** 2).mean()
    loss.backward()
    opt.step()
    if step == 0:
        loss0 = float(loss.detach())
final = float(loss.detach())
assert final < loss0 * 0.2, (loss0, final)
opt.unregister_window()
print(f"WINPUT_OPT_OK rank={rank} loss {loss0:.4f}->{final:.4f}", flush=True)
"""


def test_gpu_win_put_optimizer_two_ranks_one_device():
    """End-to-end async-gossip training (DistributedWinPutOptimizer, the
    BASELINE config-4 path) with the HIP-IPC window data plane: two ranks
    share cuda:0, parameters gossip through win_put + win_update."""
    if not torch.cuda.is_available():
        pytest.skip("needs GPU")
    port = _free_port()
    procs = []
    for rank in range(2):
        env = dict(os.environ)
        env.update(
            BF_ROOT=_ROOT,
            RANK=str(rank),
            LOCAL_RANK=str(rank),
            WORLD_SIZE="2",
            MASTER_ADDR="127.0.0.1",
            MASTER_PORT=str(port),
            BLUEFOG_BACKEND="gloo",
            HSA_ENABLE_IPC_MODE_LEGACY="0",
        )
        procs.append(
            subprocess.Popen(
                [sys.executable, "-c", _OPT_WORKER],
                env=env,
                stdout=subprocess.PIPE,
                stderr=subprocess.STDOUT,
            )
        )
    for rank, p in enumerate(procs):
        try:
            out, _ = p.communicate(timeout=300)
        except subprocess.TimeoutExpired:
            p.kill()
            out, _ = p.communicate()
        text = out.decode(errors="replace")
        assert p.returncode == 0, f"rank {rank} failed:\n{text}"
        assert f"WINPUT_OPT_OK rank={rank}" in text, text
