# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""Shared op-submission plumbing.

The reference routes every op through a background C++ thread (tensor queue
-> negotiation -> controller; reference: bluefog/common/operations.cc:
453-1185). Here submission is stream-ordered:

- GPU: RCCL work is posted immediately on ProcessGroupNCCL's comm streams;
  the post-communication kernel runs on a dedicated high-priority side
  stream that waits on the comm work via hipEvents; the returned handle
  carries the side stream's completion event. The host never blocks.
- CPU (gloo): works are posted async; the post-op math runs lazily at
  ``synchronize``.
"""

import threading
from typing import Callable, List, Optional, Sequence, Tuple

import torch
import torch.distributed as dist

from bluefog_amd.ops.context import ctx
from bluefog_amd.ops.handles import handle_manager
from bluefog_amd.utils.timeline import timeline

_name_counters = {}
_name_lock = threading.Lock()


def auto_name(op: str, name: Optional[str]) -> str:
    if name:
        return f"{op}.{name}"
    with _name_lock:
        c = _name_counters.get(op, 0)
        _name_counters[op] = c + 1
    return f"{op}.noname.{c}"


def wait_if_suspended() -> None:
    """Honor bf.suspend()/resume(): while suspended, op submission blocks
    (reference analog: suspending the background comm thread,
    operations.cc:1392-1400 — ops queue until resume). Event-backed: no
    polling while suspended, resume wakes submitters immediately."""
    ctx().wait_until_running()


def submit(
    name: str,
    works: List,
    finalize: Callable[[], torch.Tensor],
    device: torch.device,
    keep_alive: Sequence[torch.Tensor] = (),
    nbytes: Optional[int] = None,
    fingerprint: str = "",
    fp_detail: str = "",
) -> int:
    """Create a handle for posted works + a post-op callback and, on GPU,
    run the callback now on the side stream (stream-ordered)."""
    h = handle_manager().allocate(name)
    from bluefog_amd.utils import metrics

    metrics.record_submit(name, nbytes)
    from bluefog_amd.ops.consistency import checker

    checker().record(name, nbytes, fingerprint, fp_detail)
    tl = timeline()
    tl.start_activity(name, "COMMUNICATE")
    if device.type == "cuda":
        side = ctx().side_stream()
        cur = torch.cuda.current_stream()
        side.wait_stream(cur)
        with torch.cuda.stream(side):
            if tl.enabled:
                # hipEvent pair brackets the GPU execution window (comm
                # wait + post-kernel); the timeline poller stamps it when
                # it retires — execution time, not posting time
                sev = torch.cuda.Event(enable_timing=True)
                sev.record(side)
            for w in works:
                w.wait()  # stream-level dependency only; host does not block
            result = finalize()
            ev = torch.cuda.Event(enable_timing=tl.enabled)
            ev.record(side)
        if tl.enabled:
            tl.gpu_span(name, "COMMUNICATE_GPU", device, sev, ev)
        for t in keep_alive:
            if t is not None and t.is_cuda:
                t.record_stream(side)
        if result is not None and result.is_cuda:
            result.record_stream(cur)
        h.event = ev
        h.result = result
    else:
        h.works = works
        h.finalize = finalize
    return h.id


def poll(handle: int) -> bool:
    """Whether the op behind ``handle`` has completed."""
    return handle_manager().poll(handle)


def synchronize(handle: int) -> torch.Tensor:
    """Retrieve the op's output. On GPU this only *stream-orders* the caller
    behind the result (host-nonblocking); on CPU it waits for the works and
    runs the post-op math."""
    return handle_manager().synchronize(handle)


def wait(handle: int) -> torch.Tensor:
    """Like synchronize, but fully blocks the host until retirement."""
    h = handle_manager().get(handle)
    try:
        out = h.wait_host()
    finally:
        handle_manager().release(handle)
    return out


def batch_p2p(p2p_ops: List[dist.P2POp]) -> List:
    if not p2p_ops:
        return []
    return list(dist.batch_isend_irecv(p2p_ops))


def exchange_first_dims(
    d0: int, src_ranks: Sequence[int], dst_ranks: Sequence[int]
) -> List[int]:
    """Blocking exchange of first-dimension sizes with p2p neighbors over
    the CPU/gloo lane (reference analog: mpi_context.cc:675-686 pre-exchange
    for output allocation). Returns sizes in ``src_ranks`` order."""
    send_t = torch.tensor([d0], dtype=torch.int64)
    recvs = [torch.zeros(1, dtype=torch.int64) for _ in src_ranks]
    ops = [dist.P2POp(dist.isend, send_t, dst) for dst in dst_ranks]
    ops += [dist.P2POp(dist.irecv, r, src) for src, r in zip(src_ranks, recvs)]
    for w in batch_p2p(ops):
        w.wait()
    return [int(r.item()) for r in recvs]


def check_src_dst_consistency(
    src_ranks: Sequence[int], dst_ranks: Sequence[int], op: str
) -> None:
    """Debug coordinator: verify every rank's dynamic send set matches the
    receivers' recv sets (reference analog: the boolean-matrix allgather of
    mpi_controller.cc:364-417). Collective over gloo; only runs when
    ``enable_topo_check`` is requested."""
    size = ctx().size()
    row = torch.zeros(2 * size, dtype=torch.uint8)
    for d in dst_ranks:
        row[d] = 1
    for s in src_ranks:
        row[size + s] = 1
    rows = [torch.zeros_like(row) for _ in range(size)]
    dist.all_gather(rows, row)
    send = torch.stack(rows)[:, :size]
    recv = torch.stack(rows)[:, size:]
    if not torch.equal(send, recv.T):
        raise ValueError(
            f"bluefog_amd {op}: dynamic topology mismatch — some rank's "
            "dst_ranks is not mirrored by the destination's src_ranks. "
            "Fix the generator or pass enable_topo_check=False only when "
            "the schedule is known-consistent."
        )


def resolve_recv_weights(
    self_weight: Optional[float],
    src_weights,
    dst_weights,
) -> Tuple[float, dict, dict, bool, bool]:
    """Weight-resolution semantics of the reference Python layer
    (mpi_ops.py:482-535): returns (self_weight, src_weights{rank: w},
    dst_weights{rank: w}, dynamic_enabled, dst_weighting_enabled)."""
    import numpy as np

    c = ctx()
    if dst_weights is None:
        dst_weights_d = {r: 1.0 for r in c.out_neighbor_ranks()}
        dynamic_enabled = False
        dst_weighting_enabled = False
    else:
        if len(set(dst_weights)) != len(dst_weights):
            raise ValueError("dst_weights must not list the same rank twice.")
        if self_weight is None or src_weights is None:
            raise ValueError(
                "Dynamic topology (dst_weights given) also requires self_weight "
                "and src_weights."
            )
        dynamic_enabled = True
        if isinstance(dst_weights, (list, tuple)):
            dst_weights_d = {int(d): 1.0 for d in dst_weights}
        else:
            dst_weights_d = {int(d): float(w) for d, w in dst_weights.items()}
        if c.rank() in dst_weights_d:
            raise ValueError(
                "The key of dst_weights should only contain other ranks "
                "(self-rank is not allowed; use self_weight)."
            )
        dst_weighting_enabled = not np.allclose(list(dst_weights_d.values()), 1.0)

    if self_weight is None and src_weights is None:
        # static graph defaults
        if c.is_topo_weighted():
            from bluefog_amd.parallel.topology import GetRecvWeights

            self_weight, src_weights = GetRecvWeights(c.load_topology(), c.rank())
        else:
            n = len(c.in_neighbor_ranks())
            w = 1.0 / (n + 1)
            self_weight = w
            src_weights = {r: w for r in c.in_neighbor_ranks()}
    elif self_weight is not None and src_weights is not None:
        if not isinstance(src_weights, dict):
            raise ValueError(
                "src_weights must be a dict mapping in-neighbor rank -> weight."
            )
        if not isinstance(self_weight, float):
            raise ValueError("self_weight must be a float.")
        if c.rank() in src_weights:
            raise ValueError(
                "The key of src_weights should only contain other ranks "
                "(self-rank is not allowed; use self_weight)."
            )
        if not dynamic_enabled and not set(src_weights.keys()).issubset(
            set(c.in_neighbor_ranks())
        ):
            raise ValueError(
                "src_weights may only name static in-neighbor ranks; use dynamic "
                "mode (pass dst_weights) for arbitrary peers."
            )
    else:
        raise ValueError(
            "self_weight and src_weights must be given together (or both omitted)"
        )
    return self_weight, dict(src_weights), dst_weights_d, dynamic_enabled, dst_weighting_enabled
