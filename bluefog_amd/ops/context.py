# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""Global runtime context: rendezvous, process groups, topology state.

Replaces the reference's ctypes core (reference: bluefog/common/basics.py:
37-568 and the C++ init path, operations.cc:1189-1314) with a torchrun-style
env rendezvous. One process per GPU; ``torch.distributed`` is initialized
with the combined ``cpu:gloo,cuda:nccl`` backend so CPU tensors ride gloo
and CUDA tensors ride RCCL over xGMI from the same default group. A
:class:`~bluefog_amd.ops.store_util.ControlStore` on rank 0's TCPStore is
the control plane (window registry, distributed mutex, versions).

There is no background communication thread: RCCL work is stream-ordered
(see DESIGN.md). ``suspend()``/``resume()`` therefore only gate op
submission, which is all the reference API promises.
"""

import atexit
import datetime
import os
import socket
import threading
from typing import Callable, Dict, List, Optional

import torch
import torch.distributed as dist

from bluefog_amd.graph import DiGraph, as_digraph
from bluefog_amd.parallel import topology as topology_util
from bluefog_amd.utils.logging import get_logger

logger = get_logger()

_NULL = object()


class BlueFogContext:
    """Singleton holding all distributed state for this process."""

    def __init__(self):
        self._initialized = False
        self._rank = -1
        self._size = -1
        self._local_rank = -1
        self._local_size = -1
        self._machine_rank = -1
        self._machine_size = -1
        self._is_homogeneous = True
        self._device: Optional[torch.device] = None
        self._store = None  # ControlStore
        self._topology: Optional[DiGraph] = None
        self._machine_topology: Optional[DiGraph] = None
        self._is_topo_weighted = False
        self._is_machine_topo_weighted = False
        self._in_neighbor_ranks: List[int] = []
        self._out_neighbor_ranks: List[int] = []
        self._local_group = None
        self._cross_group = None
        self._local_groups_built = False
        self._machine_rank_lists: List[List[int]] = []
        # event semantics: set = running, cleared = suspended (submitters
        # block in Event.wait instead of spin-polling)
        self._running = threading.Event()
        self._running.set()
        self._skip_negotiate = True  # program-order contract is the default
        self._owns_process_group = False
        # window registry lives in ops.window; it registers itself here so
        # set_topology can refuse while windows exist (reference parity,
        # basics.py:405-416).
        self.window_registry = None
        self._side_streams: Dict[int, torch.cuda.Stream] = {}
        self._local_comm_streams: Dict[int, torch.cuda.Stream] = {}

    # ------------------------------------------------------------------
    # lifecycle
    # ------------------------------------------------------------------
    def init(
        self,
        topology_fn: Optional[Callable[[int], DiGraph]] = None,
        is_weighted: bool = False,
    ) -> None:
        """Initialize the framework.

        Reads ``RANK``/``WORLD_SIZE``/``LOCAL_RANK``/``MASTER_ADDR``/
        ``MASTER_PORT`` from the environment (as set by ``bfrun`` or
        ``torchrun``); absent any of them, runs single-process. Sets the
        default virtual topology to ``ExponentialGraph(size)`` unless
        ``topology_fn`` is given (reference semantics: basics.py:49-70).
        """
        if self._initialized:
            logger.warning("bluefog_amd already initialized; ignoring second init()")
            return

        rank = int(os.environ.get("RANK", "0"))
        size = int(os.environ.get("WORLD_SIZE", "1"))
        local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", str(_pick_free_port() if size == 1 else 29500))

        use_cuda = torch.cuda.is_available()
        if use_cuda:
            torch.cuda.set_device(local_rank % torch.cuda.device_count())
            self._device = torch.device("cuda", torch.cuda.current_device())
        else:
            self._device = torch.device("cpu")

        if not dist.is_initialized():
            # BLUEFOG_BACKEND overrides (e.g. "gloo" to run several ranks on
            # one GPU in tests — RCCL forbids two ranks on one device)
            backend = os.environ.get(
                "BLUEFOG_BACKEND", "cpu:gloo,cuda:nccl" if use_cuda else "gloo"
            )
            kwargs = {}
            if use_cuda and "nccl" in backend:
                kwargs["device_id"] = self._device
            dist.init_process_group(
                backend=backend,
                rank=rank,
                world_size=size,
                timeout=datetime.timedelta(
                    seconds=int(os.environ.get("BLUEFOG_INIT_TIMEOUT", "300"))
                ),
                **kwargs,
            )
            self._owns_process_group = True

        self._rank = dist.get_rank()
        self._size = dist.get_world_size()

        from bluefog_amd.ops.store_util import ControlStore

        self._store = ControlStore(_default_store(), self._rank, self._size)

        self._detect_machines(local_rank)
        self._initialized = True

        if topology_fn:
            topo = topology_fn(self._size)
        else:
            topo = topology_util.ExponentialGraph(self._size)
        self.set_topology(topo, is_weighted)
        atexit.register(self.shutdown)

    def _detect_machines(self, local_rank_hint: int) -> None:
        """Group ranks into machines. ``BLUEFOG_NODES_PER_MACHINE`` fakes
        multiple machines inside one node for hierarchical tests (reference:
        mpi_context.cc:320-337); otherwise group by hostname."""
        npm_env = os.environ.get("BLUEFOG_NODES_PER_MACHINE")
        if npm_env is not None:
            npm = int(npm_env)
            assert npm > 0 and self._size % npm == 0, (
                "BLUEFOG_NODES_PER_MACHINE must divide the world size"
            )
            self._local_size = npm
            self._local_rank = self._rank % npm
            self._machine_rank = self._rank // npm
            self._machine_size = self._size // npm
            self._is_homogeneous = True
            self._machine_rank_lists = [
                list(range(m * npm, (m + 1) * npm)) for m in range(self._machine_size)
            ]
            return
        if self._size == 1:
            self._local_size = 1
            self._local_rank = 0
            self._machine_rank = 0
            self._machine_size = 1
            self._machine_rank_lists = [[0]]
            return
        # hostname census through the store
        host = socket.gethostname()
        self._store.set(f"host/{self._rank}", host.encode())
        self._store.wait([f"host/{r}" for r in range(self._size)])
        hosts = [self._store.get(f"host/{r}").decode() for r in range(self._size)]
        unique_hosts = sorted(set(hosts), key=hosts.index)
        by_host: Dict[str, List[int]] = {h: [] for h in unique_hosts}
        for r, h in enumerate(hosts):
            by_host[h].append(r)
        self._machine_rank_lists = [by_host[h] for h in unique_hosts]
        my_machine = hosts[self._rank]
        self._machine_rank = unique_hosts.index(my_machine)
        self._machine_size = len(unique_hosts)
        self._local_size = len(by_host[my_machine])
        self._local_rank = by_host[my_machine].index(self._rank)
        sizes = {len(v) for v in by_host.values()}
        self._is_homogeneous = len(sizes) == 1
        del local_rank_hint

    def shutdown(self) -> None:
        if not self._initialized:
            return
        try:
            if self.window_registry is not None:
                self.window_registry.free_all()
        except Exception:  # pragma: no cover - teardown best effort
            pass
        if self._owns_process_group and dist.is_initialized():
            try:
                dist.destroy_process_group()
            except Exception:  # pragma: no cover
                pass
        self._initialized = False
        self._topology = None
        self._machine_topology = None
        self._local_group = None
        self._cross_group = None
        self._local_groups_built = False
        self._side_streams.clear()
        self._local_comm_streams.clear()

    # ------------------------------------------------------------------
    # accessors
    # ------------------------------------------------------------------
    def _require_init(self) -> None:
        if not self._initialized:
            raise ValueError("BlueFog has not been initialized; use bf.init().")

    def rank(self) -> int:
        self._require_init()
        return self._rank

    def size(self) -> int:
        self._require_init()
        return self._size

    def local_rank(self) -> int:
        self._require_init()
        return self._local_rank

    def local_size(self) -> int:
        self._require_init()
        return self._local_size

    def machine_rank(self) -> int:
        self._require_init()
        assert self.is_homogeneous(), "machine faking requires every machine to hold the same number of ranks"
        return self._machine_rank

    def machine_size(self) -> int:
        self._require_init()
        assert self.is_homogeneous(), "machine faking requires every machine to hold the same number of ranks"
        return self._machine_size

    def is_homogeneous(self) -> bool:
        self._require_init()
        return self._is_homogeneous

    def is_initialized(self) -> bool:
        return self._initialized

    @property
    def store(self):
        self._require_init()
        return self._store

    @property
    def device(self) -> torch.device:
        self._require_init()
        return self._device

    def suspend(self) -> None:
        self._running.clear()

    def resume(self) -> None:
        self._running.set()

    @property
    def suspended(self) -> bool:
        return not self._running.is_set()

    def wait_until_running(self) -> None:
        self._running.wait()

    def set_skip_negotiate_stage(self, value: bool) -> None:
        self._skip_negotiate = bool(value)

    def get_skip_negotiate_stage(self) -> bool:
        return self._skip_negotiate

    # ------------------------------------------------------------------
    # topology
    # ------------------------------------------------------------------
    def set_topology(self, topology=None, is_weighted: bool = False) -> bool:
        self._require_init()
        if topology is None:
            topology = topology_util.ExponentialGraph(self._size)
            if self._local_rank == 0:
                logger.info(
                    "Topology is not specified. Default Exponential Two topology is used."
                )
        topology = as_digraph(topology)
        if topology.number_of_nodes() != self._size:
            raise TypeError(
                "topology must be a DiGraph with the same number of nodes as bf.size()."
            )
        if topology_util.IsTopologyEquivalent(topology, self._topology):
            logger.debug("Topology to set is the same as old one. Skip the setting.")
            return True
        if self.window_registry is not None and self.window_registry.names():
            if self._local_rank == 0:
                logger.error(
                    "Cannot set topology while windows are registered. Call "
                    "bf.win_free() first, then set the topology."
                )
            return False
        self._topology = topology
        self._is_topo_weighted = is_weighted
        me = self._rank
        self._in_neighbor_ranks = sorted(
            r for r in topology.predecessors(me) if r != me
        )
        self._out_neighbor_ranks = sorted(
            r for r in topology.successors(me) if r != me
        )
        return True

    def set_machine_topology(self, topology, is_weighted: bool = False) -> bool:
        self._require_init()
        if topology is None:
            raise ValueError("Machine topology shall not be None.")
        topology = as_digraph(topology)
        if topology.number_of_nodes() != self.machine_size():
            raise TypeError(
                "machine topology must have the same number of nodes as bf.machine_size()."
            )
        assert self.is_homogeneous(), "machine faking requires every machine to hold the same number of ranks"
        if topology_util.IsTopologyEquivalent(topology, self._machine_topology):
            logger.debug("Machine topology to set is the same as old one. Skip.")
            return True
        self._machine_topology = topology
        self._is_machine_topo_weighted = is_weighted
        return True

    def load_topology(self) -> Optional[DiGraph]:
        self._require_init()
        return self._topology

    def load_machine_topology(self) -> Optional[DiGraph]:
        self._require_init()
        return self._machine_topology

    def is_topo_weighted(self) -> bool:
        self._require_init()
        return self._is_topo_weighted

    def is_machine_topo_weighted(self) -> bool:
        self._require_init()
        return self._is_machine_topo_weighted

    def in_neighbor_ranks(self) -> List[int]:
        self._require_init()
        return list(self._in_neighbor_ranks)

    def out_neighbor_ranks(self) -> List[int]:
        self._require_init()
        return list(self._out_neighbor_ranks)

    def in_neighbor_machine_ranks(self) -> List[int]:
        self._require_init()
        if self._machine_topology is None:
            return []
        m = self.machine_rank()
        return [r for r in self._machine_topology.predecessors(m) if r != m]

    def out_neighbor_machine_ranks(self) -> List[int]:
        self._require_init()
        if self._machine_topology is None:
            return []
        m = self.machine_rank()
        return [r for r in self._machine_topology.successors(m) if r != m]

    # ------------------------------------------------------------------
    # process groups & streams
    # ------------------------------------------------------------------
    def machine_rank_list(self, machine: int) -> List[int]:
        """Global ranks of a machine, in local-rank order (leader first)."""
        self._require_init()
        return list(self._machine_rank_lists[machine])

    def ensure_local_groups(self) -> None:
        """Build the per-machine ("local") and same-local-rank ("cross")
        process groups used by hierarchical ops. Collective: every rank must
        reach this before any hierarchical op (reference analog:
        mpi_context.cc:320-344)."""
        self._require_init()
        if self._local_groups_built:
            return
        assert self._is_homogeneous, "hierarchical groups need a homogeneous placement"
        local_groups = [dist.new_group(ranks=r) for r in self._machine_rank_lists]
        self._local_group = local_groups[self._machine_rank]
        nls = self._local_size
        cross_lists = [
            [m_ranks[i] for m_ranks in self._machine_rank_lists]
            for i in range(nls)
        ]
        cross_groups = [dist.new_group(ranks=r) for r in cross_lists]
        self._cross_group = cross_groups[self._local_rank]
        self._local_groups_built = True

    @property
    def local_group(self):
        self.ensure_local_groups()
        return self._local_group

    @property
    def cross_group(self):
        self.ensure_local_groups()
        return self._cross_group

    def side_stream(self) -> torch.cuda.Stream:
        """Per-device side HIP stream for post-communication kernels
        (weighted average, fused optimizer step) overlapped with compute."""
        dev = torch.cuda.current_device()
        if dev not in self._side_streams:
            self._side_streams[dev] = torch.cuda.Stream(priority=-1)
        return self._side_streams[dev]


_context = BlueFogContext()


def ctx() -> BlueFogContext:
    return _context


def _pick_free_port() -> int:
    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _default_store():
    """The c10d store backing the default process group."""
    try:
        store = dist.distributed_c10d._get_default_store()
        if store is not None:
            return store
    except Exception:
        pass
    # Fallback: a dedicated TCPStore next to the rendezvous port.
    rank = dist.get_rank()
    size = dist.get_world_size()
    host = os.environ.get("MASTER_ADDR", "127.0.0.1")
    port = int(os.environ.get("BLUEFOG_STORE_PORT", int(os.environ["MASTER_PORT"]) + 1))
    return dist.TCPStore(host, port, size, rank == 0)
