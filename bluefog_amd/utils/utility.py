# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""Parameter/optimizer synchronization helpers (reference analog:
bluefog/torch/utility.py:26-216). These are the checkpoint/resume story of
the framework: load a standard torch ``state_dict``, then broadcast it so
every rank restarts identical."""

import collections
from typing import Union

import torch

__all__ = [
    "broadcast_parameters",
    "allreduce_parameters",
    "broadcast_optimizer_state",
]


def _normalize_params(params):
    if isinstance(params, dict):
        return sorted(params.items())
    if isinstance(params, list):
        return [p if isinstance(p, tuple) else (None, p) for p in params]
    raise ValueError("invalid params of type: %s" % type(params))


def broadcast_parameters(params, root_rank: int) -> None:
    """Broadcast ``model.state_dict()`` / ``named_parameters()`` /
    ``parameters()`` from ``root_rank`` to all ranks, in place."""
    from bluefog_amd.ops import collective, engine

    handles = []
    for name, p in _normalize_params(params):
        handles.append(collective.broadcast_nonblocking_(p, root_rank, name))
    for handle in handles:
        engine.synchronize(handle)


def allreduce_parameters(params) -> None:
    """Average the given parameters over all ranks, in place — pulls a
    decentralized run back to consensus."""
    from bluefog_amd.ops import collective, engine

    handles = []
    for name, p in _normalize_params(params):
        handles.append(collective.allreduce_nonblocking_(p, average=True, name=name))
    for handle in handles:
        engine.synchronize(handle)


def broadcast_optimizer_state(optimizer: torch.optim.Optimizer, root_rank: int) -> None:
    """Broadcast an optimizer's state dict from ``root_rank``; scalar
    hyper-state is tensorized for transport and cast back afterwards."""
    from bluefog_amd.ops import collective

    if isinstance(optimizer, torch.optim.LBFGS):
        raise ValueError("cannot broadcast torch.optim.LBFGS state")

    state_dict = optimizer.state_dict()

    # Initialize fresh optimizers so the state exists to broadcast.
    if not state_dict["state"]:
        for group in optimizer.param_groups:
            for p in group["params"]:
                if p.requires_grad and p.grad is None:
                    p.grad = p.data.new_zeros(p.size())
        # call the *base* step when the optimizer is one of our distributed
        # wrappers (their step() would launch collective comm)
        base_step = getattr(optimizer, "_bluefog_base_step", None)
        if base_step is not None:
            base_step()
        else:
            optimizer.step()
        state_dict = optimizer.state_dict()
    if not state_dict["state"]:
        return  # stateless optimizer

    occurrences = collections.defaultdict(int)
    callbacks = []

    def _from_tensor(value, t: torch.Tensor):
        # cast a broadcast 1-element tensor back to the python scalar type
        if isinstance(value, bool):
            return bool(t.item())
        if isinstance(value, int):
            return int(t.item())
        if isinstance(value, float):
            return float(t.item())
        return t

    for pid, pstate in sorted(state_dict["state"].items()):
        for key, value in sorted(pstate.items()):
            occurrences[key] += 1
            name = f"optstate.{key}.{occurrences[key]}"
            if torch.is_tensor(value):
                collective.broadcast_(value, root_rank, name)
            elif value is not None and isinstance(value, (bool, int, float)):
                t = torch.tensor([float(value)], dtype=torch.float64)
                collective.broadcast_(t, root_rank, name)
                callbacks.append((pstate, key, value, t))

    for pstate, key, value, t in callbacks:
        pstate[key] = _from_tensor(value, t)

    optimizer.load_state_dict(state_dict)
