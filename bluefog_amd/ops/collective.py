# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""Global collective ops: allreduce / broadcast / allgather / barrier.

Reference analog: the MPI_Allreduce / MPI_Bcast / MPI_Allgatherv and
ncclAllReduce / ncclBcast / ncclAllGather call sites listed in SURVEY.md
§2.4 (mpi_controller.cc:136-213, nccl_controller.cc:386-500). Here they map
1:1 onto torch.distributed collectives — RCCL over xGMI for CUDA tensors,
gloo for CPU tensors — with the average division fused in the post-op.
"""

from typing import Optional

import torch
import torch.distributed as dist

from bluefog_amd.ops import engine
from bluefog_amd.ops.context import ctx

__all__ = [
    "allreduce",
    "allreduce_nonblocking",
    "allreduce_",
    "allreduce_nonblocking_",
    "broadcast",
    "broadcast_nonblocking",
    "broadcast_",
    "broadcast_nonblocking_",
    "allgather",
    "allgather_nonblocking",
    "barrier",
]


def _allreduce_impl(
    tensor: torch.Tensor,
    output: torch.Tensor,
    average: bool,
    is_hierarchical_local: bool,
    name: Optional[str],
) -> int:
    engine.wait_if_suspended()
    c = ctx()
    if is_hierarchical_local:
        assert c.is_homogeneous(), (
            "hierarchical local allreduce needs a homogeneous placement"
        )
        group = c.local_group
        group_size = c.local_size()
    else:
        group = None
        group_size = c.size()
    if output.data_ptr() != tensor.data_ptr():
        output.copy_(tensor)
    work = dist.all_reduce(output, op=dist.ReduceOp.SUM, group=group, async_op=True)

    def finalize():
        if average:
            output.div_(group_size)
        return output

    return engine.submit(
        engine.auto_name("allreduce", name),
        [work],
        finalize,
        output.device,
        keep_alive=(output,),
        nbytes=output.numel() * output.element_size(),
    )


def allreduce(
    tensor: torch.Tensor,
    average: bool = True,
    is_hierarchical_local: bool = False,
    name: Optional[str] = None,
) -> torch.Tensor:
    """Sum (or average) over all ranks; the input is not modified."""
    handle = allreduce_nonblocking(tensor, average, is_hierarchical_local, name)
    return engine.synchronize(handle)


def allreduce_nonblocking(
    tensor: torch.Tensor,
    average: bool = True,
    is_hierarchical_local: bool = False,
    name: Optional[str] = None,
) -> int:
    output = tensor.detach().clone()
    return _allreduce_impl(tensor, output, average, is_hierarchical_local, name)


def allreduce_(
    tensor: torch.Tensor,
    average: bool = True,
    is_hierarchical_local: bool = False,
    name: Optional[str] = None,
) -> torch.Tensor:
    """In-place allreduce."""
    handle = allreduce_nonblocking_(tensor, average, is_hierarchical_local, name)
    return engine.synchronize(handle)


def allreduce_nonblocking_(
    tensor: torch.Tensor,
    average: bool = True,
    is_hierarchical_local: bool = False,
    name: Optional[str] = None,
) -> int:
    return _allreduce_impl(tensor, tensor, average, is_hierarchical_local, name)


def broadcast(
    tensor: torch.Tensor, root_rank: int, name: Optional[str] = None
) -> torch.Tensor:
    handle = broadcast_nonblocking(tensor, root_rank, name)
    return engine.synchronize(handle)


def broadcast_nonblocking(
    tensor: torch.Tensor, root_rank: int, name: Optional[str] = None
) -> int:
    engine.wait_if_suspended()
    output = tensor.detach().clone()
    work = dist.broadcast(output, src=root_rank, async_op=True)
    return engine.submit(
        engine.auto_name("broadcast", name),
        [work],
        lambda: output,
        output.device,
        keep_alive=(output,),
    )


def broadcast_(tensor: torch.Tensor, root_rank: int, name: Optional[str] = None):
    handle = broadcast_nonblocking_(tensor, root_rank, name)
    return engine.synchronize(handle)


def broadcast_nonblocking_(
    tensor: torch.Tensor, root_rank: int, name: Optional[str] = None
) -> int:
    engine.wait_if_suspended()
    work = dist.broadcast(tensor, src=root_rank, async_op=True)
    return engine.submit(
        engine.auto_name("broadcast", name),
        [work],
        lambda: tensor,
        tensor.device,
        keep_alive=(tensor,),
    )


def allgather(tensor: torch.Tensor, name: Optional[str] = None) -> torch.Tensor:
    handle = allgather_nonblocking(tensor, name)
    return engine.synchronize(handle)


def allgather_nonblocking(tensor: torch.Tensor, name: Optional[str] = None) -> int:
    """Concatenate the tensor from every rank along dim 0; first dims may
    differ (reference allows ragged first dims, MPI_Allgatherv)."""
    engine.wait_if_suspended()
    c = ctx()
    size = c.size()
    t = tensor.detach()
    if not t.is_contiguous():
        t = t.contiguous()
    d0 = t.shape[0] if t.dim() else 1
    # first-dim census over the CPU lane
    dims = [torch.zeros(1, dtype=torch.int64) for _ in range(size)]
    dist.all_gather(dims, torch.tensor([d0], dtype=torch.int64))
    d0s = [int(x.item()) for x in dims]
    rest = list(t.shape[1:])
    if len(set(d0s)) == 1:
        output = t.new_empty([d0 * size] + rest)
        chunks = [output.narrow(0, i * d0, d0) for i in range(size)]
        work = dist.all_gather(chunks, t, async_op=True)
        return engine.submit(
            engine.auto_name("allgather", name),
            [work],
            lambda: output,
            output.device,
            keep_alive=(output, t),
        )
    # ragged: pad to max, gather, then compact
    dmax = max(d0s)
    padded = t.new_zeros([dmax] + rest)
    if d0:
        padded.narrow(0, 0, d0).copy_(t)
    gathered = [t.new_empty([dmax] + rest) for _ in range(size)]
    work = dist.all_gather(gathered, padded, async_op=True)
    output = t.new_empty([sum(d0s)] + rest)

    def finalize():
        off = 0
        for i, n in enumerate(d0s):
            if n:
                output.narrow(0, off, n).copy_(gathered[i].narrow(0, 0, n))
            off += n
        return output

    return engine.submit(
        engine.auto_name("allgather", name),
        [work],
        finalize,
        output.device,
        keep_alive=[output, padded] + gathered,
    )


def barrier() -> None:
    """Block until every rank arrives (a real barrier, not the reference's
    negotiated allreduce workaround, mpi_ops.py:996-1005)."""
    ctx()._require_init()
    dist.barrier()
    if torch.cuda.is_available() and torch.cuda.is_initialized():
        torch.cuda.synchronize()
