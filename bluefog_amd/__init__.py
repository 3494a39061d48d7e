# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""bluefog_amd — an MI355X-native decentralized deep-learning training
framework with the capabilities of BlueFog (see DESIGN.md / SURVEY.md).

Use exactly like the reference's ``bluefog.torch``::

    import bluefog_amd as bf          # or: import bluefog_amd.torch as bf
    bf.init()
    bf.set_topology(bf.ExponentialTwoGraph(bf.size()))
    avg = bf.neighbor_allreduce(tensor)

Public surface mirrors the reference export list
(/root/reference/bluefog/torch/__init__.py:21-110).
"""

__version__ = "0.1.0"

from bluefog_amd.graph import DiGraph

from bluefog_amd.ops.context import ctx as _ctx


# ---------------------------------------------------------------------------
# lifecycle / introspection (reference basics.py)
# ---------------------------------------------------------------------------


def init(topology_fn=None, is_weighted: bool = False):
    """Initialize bluefog_amd (rendezvous + default Exponential-2 topology)."""
    _ctx().init(topology_fn, is_weighted)
    from bluefog_amd.utils.timeline import maybe_init_from_env

    maybe_init_from_env(_ctx().rank())
    from bluefog_amd.utils import metrics as _metrics

    _metrics.maybe_start_from_env(_ctx().rank())
    # make the window registry known to the context for topology guards
    from bluefog_amd.ops import window as _window

    _window.registry()


def start_metrics_server(port: int):
    """Serve Prometheus metrics for this rank on ``port`` (see
    bluefog_amd/utils/metrics.py)."""
    from bluefog_amd.utils import metrics as _metrics

    _metrics.start_server(port)


def shutdown():
    _ctx().shutdown()


def size() -> int:
    return _ctx().size()


def local_size() -> int:
    return _ctx().local_size()


def rank() -> int:
    return _ctx().rank()


def local_rank() -> int:
    return _ctx().local_rank()


def machine_size() -> int:
    return _ctx().machine_size()


def machine_rank() -> int:
    return _ctx().machine_rank()


def is_homogeneous() -> bool:
    return _ctx().is_homogeneous()


def load_topology():
    return _ctx().load_topology()


def set_topology(topology=None, is_weighted: bool = False) -> bool:
    return _ctx().set_topology(topology, is_weighted)


def load_machine_topology():
    return _ctx().load_machine_topology()


def set_machine_topology(topology, is_weighted: bool = False) -> bool:
    return _ctx().set_machine_topology(topology, is_weighted)


def is_topo_weighted() -> bool:
    return _ctx().is_topo_weighted()


def is_machine_topo_weighted() -> bool:
    return _ctx().is_machine_topo_weighted()


def in_neighbor_ranks():
    return _ctx().in_neighbor_ranks()


def out_neighbor_ranks():
    return _ctx().out_neighbor_ranks()


def in_neighbor_machine_ranks():
    return _ctx().in_neighbor_machine_ranks()


def out_neighbor_machine_ranks():
    return _ctx().out_neighbor_machine_ranks()


def suspend():
    _ctx().suspend()


def resume():
    _ctx().resume()


def set_skip_negotiate_stage(value: bool) -> None:
    _ctx().set_skip_negotiate_stage(value)


def get_skip_negotiate_stage() -> bool:
    return _ctx().get_skip_negotiate_stage()


def mpi_threads_supported() -> bool:
    """No MPI in this framework; the control plane is the TCP store and the
    data plane is RCCL/xGMI. Kept for API compatibility."""
    return False


def unified_mpi_window_model_supported() -> bool:
    """Windows here are HIP-IPC peer memory, which is always 'unified'."""
    return True


def nccl_built() -> bool:
    """RCCL (the ROCm NCCL) is the only GPU backend of this framework."""
    import torch

    return torch.distributed.is_nccl_available()


def rccl_built() -> bool:
    return nccl_built()


# ---------------------------------------------------------------------------
# timeline
# ---------------------------------------------------------------------------


def check_extension(ext_name: str = "bluefog_amd._C", pkg_path=None, *args) -> None:
    """Raise ImportError if the native gfx950 extension is not built
    (reference analog: bluefog/common/util.py:47-52, called at package
    import to fail early). Extra args are accepted for signature parity."""
    from bluefog_amd.ops import hip_ext

    if not hip_ext.has_extension():
        raise ImportError(
            f"Extension {ext_name} has not been built. Run "
            "`PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace`."
        )


def timeline_start_activity(tensor_name: str, activity_name: str) -> bool:
    from bluefog_amd.utils.timeline import timeline

    return timeline().start_activity(tensor_name, activity_name)


def timeline_end_activity(tensor_name: str) -> bool:
    from bluefog_amd.utils.timeline import timeline

    return timeline().end_activity(tensor_name)


from contextlib import contextmanager as _contextmanager


@_contextmanager
def timeline_context(tensor_name: str, activity_name: str):
    timeline_start_activity(tensor_name, activity_name)
    try:
        yield
    finally:
        timeline_end_activity(tensor_name)


# ---------------------------------------------------------------------------
# ops
# ---------------------------------------------------------------------------

from bluefog_amd.ops.collective import (  # noqa: E402
    allgather,
    allgather_nonblocking,
    allreduce,
    allreduce_,
    allreduce_nonblocking,
    allreduce_nonblocking_,
    barrier,
    broadcast,
    broadcast_,
    broadcast_nonblocking,
    broadcast_nonblocking_,
)
from bluefog_amd.ops.engine import poll, synchronize, wait  # noqa: E402
from bluefog_amd.ops.neighbor import (  # noqa: E402
    hierarchical_neighbor_allreduce,
    hierarchical_neighbor_allreduce_nonblocking,
    neighbor_allgather,
    neighbor_allgather_nonblocking,
    neighbor_allreduce,
    neighbor_allreduce_nonblocking,
    pair_gossip,
    pair_gossip_nonblocking,
)
from bluefog_amd.ops.window import (  # noqa: E402
    get_current_created_window_names,
    get_win_version,
    neighbor_win_accumulate,
    neighbor_win_accumulate_nonblocking,
    neighbor_win_get,
    neighbor_win_get_nonblocking,
    neighbor_win_put,
    neighbor_win_put_nonblocking,
    turn_off_win_ops_with_associated_p,
    turn_on_win_ops_with_associated_p,
    win_accumulate,
    win_accumulate_nonblocking,
    win_associated_p,
    win_create,
    win_free,
    win_get,
    win_get_nonblocking,
    win_lock,
    win_mutex,
    win_poll,
    win_put,
    win_put_nonblocking,
    win_update,
    win_update_then_collect,
    win_wait,
)

# ---------------------------------------------------------------------------
# optimizers
# ---------------------------------------------------------------------------

from bluefog_amd.optimizers import (  # noqa: E402
    CommunicationType,
    DistributedAdaptThenCombineOptimizer,
    DistributedAdaptWithCombineOptimizer,
    DistributedAllreduceOptimizer,
    DistributedGradientAllreduceOptimizer,
    DistributedHierarchicalNeighborAllreduceOptimizer,
    DistributedNeighborAllreduceOptimizer,
    DistributedPullGetOptimizer,
    DistributedPushSumOptimizer,
    DistributedWinPutOptimizer,
)

# ---------------------------------------------------------------------------
# utilities & topology library
# ---------------------------------------------------------------------------

from bluefog_amd.utils.utility import (  # noqa: E402
    allreduce_parameters,
    broadcast_optimizer_state,
    broadcast_parameters,
)

from bluefog_amd.parallel.topology import (  # noqa: E402
    ExponentialGraph,
    ExponentialTwoGraph,
    FullyConnectedGraph,
    GetDynamicOnePeerSendRecvRanks,
    GetExp2DynamicSendRecvMachineRanks,
    GetInnerOuterExpo2DynamicSendRecvRanks,
    GetInnerOuterRingDynamicSendRecvRanks,
    GetRecvWeights,
    GetSendWeights,
    IsRegularGraph,
    IsTopologyEquivalent,
    MeshGrid2DGraph,
    RingGraph,
    StarGraph,
    SymmetricExponentialGraph,
)

from bluefog_amd.parallel.topology_infer import (  # noqa: E402
    InferDestinationFromSourceRanks,
    InferSourceFromDestinationRanks,
)
