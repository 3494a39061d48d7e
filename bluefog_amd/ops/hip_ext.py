# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""Dispatch layer over the native HIP extension ``bluefog_amd._C``.

The extension carries the hand-written CDNA4 kernels (csrc/bluefog_kernels.hip) that
replace the reference's chain of torch slice arithmetic in the
post-communication callbacks (reference: bluefog/torch/mpi_ops.cc:99-164,
mpi_win_ops.cc:185-279) and the reference's lone CUDA kernel family
(cuda/cuda_kernels.cu ``scale_buffer``):

- ``weighted_combine``  out = self_w * self + sum_k w[k] * gathered[k]
- ``weighted_combine_sgd`` / ``_adam``  the same, fused with the optimizer step
- ``scale_put`` / ``accum_put``  one-pass scale-and-store into (peer) memory

On a CUDA/ROCm device the extension is REQUIRED: a missing ``_C`` raises
instead of silently falling back to torch (so a GPU run can never pass on an
eager fallback unnoticed). ``BLUEFOG_ALLOW_TORCH_FALLBACK=1`` overrides for
debugging. On CPU the torch implementations are the real path.
"""

from typing import List, Optional, Sequence

import torch

from bluefog_amd.utils.env import allow_torch_fallback

_C = None
_C_err: Optional[str] = None
try:
    from bluefog_amd import _C as _C  # type: ignore
except Exception as e:  # pragma: no cover - exercised only sans extension
    _C_err = f"{type(e).__name__}: {e}"


def has_extension() -> bool:
    return _C is not None


def _require_ext(op_name: str) -> bool:
    """True -> use native kernels; False -> torch fallback allowed."""
    if _C is not None:
        return True
    if allow_torch_fallback():
        return False
    raise RuntimeError(
        f"bluefog_amd: {op_name} on a GPU tensor requires the native HIP "
        f"extension bluefog_amd._C, which failed to import ({_C_err}). Build "
        "it with `python setup.py build_ext --inplace` "
        "(PYTORCH_ROCM_ARCH=gfx950), or set BLUEFOG_ALLOW_TORCH_FALLBACK=1 "
        "to debug with torch elementwise ops."
    )


# ---------------------------------------------------------------------------
# weighted combine: the neighbor-average primitive
# ---------------------------------------------------------------------------


def weighted_combine(
    output: torch.Tensor,
    self_tensor: torch.Tensor,
    self_weight: float,
    gathered: Optional[torch.Tensor],
    weights: Sequence[float],
) -> torch.Tensor:
    """``output = self_weight*self_tensor + sum_k weights[k]*gathered[k]``.

    ``gathered`` is the contiguous receive buffer of shape
    ``[len(weights) * n0, ...rest]`` holding one same-shaped slice per
    neighbor (the ``[self|neighbors]`` layout of the reference's fused
    buffers); ``output``/``self_tensor`` have shape ``[n0, ...rest]``.
    """
    n = len(weights)
    if output.is_cuda and _require_ext("weighted_combine"):
        _C.weighted_combine(
            output,
            self_tensor,
            float(self_weight),
            gathered if n else self_tensor,
            list(map(float, weights)),
        )
        return output
    # torch path (CPU, or explicit fallback)
    acc_dtype = (
        torch.float32
        if output.dtype in (torch.float16, torch.bfloat16)
        else output.dtype
    )
    acc = self_tensor.to(acc_dtype) * self_weight
    numel = self_tensor.numel()
    if n:
        g = gathered.reshape(n, numel)
        for k in range(n):
            acc += weights[k] * g[k].reshape(self_tensor.shape).to(acc_dtype)
    output.copy_(acc.to(output.dtype))
    return output


def weighted_combine_list(
    output: torch.Tensor,
    self_tensor: torch.Tensor,
    self_weight: float,
    neighbor_tensors: List[torch.Tensor],
    weights: Sequence[float],
) -> torch.Tensor:
    """Same math over separately-allocated neighbor tensors (window update
    path). Falls back to a loop of fused ops when buffers aren't contiguous
    slices of one allocation."""
    if (
        output.is_cuda
        and neighbor_tensors
        and _is_one_block(neighbor_tensors)
        and _require_ext("weighted_combine")
    ):
        base = _flat_view(neighbor_tensors)
        return weighted_combine(output, self_tensor, self_weight, base, weights)
    acc_dtype = (
        torch.float32
        if output.dtype in (torch.float16, torch.bfloat16)
        else output.dtype
    )
    acc = self_tensor.to(acc_dtype) * self_weight
    for w, t in zip(weights, neighbor_tensors):
        acc += w * t.to(acc_dtype)
    output.copy_(acc.to(output.dtype))
    return output


def _is_one_block(tensors: List[torch.Tensor]) -> bool:
    if len(tensors) < 2:
        return True
    first = tensors[0]
    stride = first.numel() * first.element_size()
    base = first.data_ptr()
    return all(
        t.data_ptr() == base + i * stride and t.is_contiguous()
        for i, t in enumerate(tensors)
    )


def _flat_view(tensors: List[torch.Tensor]) -> torch.Tensor:
    if len(tensors) == 1:
        return tensors[0]
    first = tensors[0]
    total0 = first.shape[0] * len(tensors) if first.dim() else len(tensors)
    shape = (total0,) + tuple(first.shape[1:])
    return first.new_empty(0).set_(
        first.untyped_storage(), first.storage_offset(), shape
    )


# ---------------------------------------------------------------------------
# fused combine + optimizer step (ATC hot path)
# ---------------------------------------------------------------------------


def weighted_combine_sgd(
    param: torch.Tensor,
    self_weight: float,
    gathered: Optional[torch.Tensor],
    weights: Sequence[float],
    grad: torch.Tensor,
    momentum_buf: Optional[torch.Tensor],
    lr: float,
    momentum: float,
    weight_decay: float,
    dampening: float = 0.0,
    nesterov: bool = False,
) -> None:
    """param = combine(param, neighbors); then SGD(momentum) step in the same
    pass over HBM."""
    if param.is_cuda and _require_ext("weighted_combine_sgd"):
        _C.weighted_combine_sgd(
            param,
            float(self_weight),
            gathered if len(weights) else param,
            list(map(float, weights)),
            grad,
            momentum_buf if momentum_buf is not None else param.new_empty(0),
            float(lr),
            float(momentum),
            float(weight_decay),
            float(dampening),
            bool(nesterov),
        )
        return
    weighted_combine(param, param.clone(), self_weight, gathered, weights)
    g = grad
    if weight_decay != 0:
        g = g.add(param, alpha=weight_decay)
    if momentum_buf is not None and momentum != 0:
        momentum_buf.mul_(momentum).add_(g, alpha=1.0 - dampening)
        # nesterov uses the weight-decay-adjusted gradient (torch semantics)
        g = g.add(momentum_buf, alpha=momentum) if nesterov else momentum_buf
    param.add_(g, alpha=-lr)


def scale(t: torch.Tensor, factor: float) -> torch.Tensor:
    """In-place buffer scaling — the reference's only CUDA kernel
    (cuda_kernels.cu:24-116). torch's ``mul_`` is already a single
    bandwidth-bound HIP kernel; the native version exists for peer-mapped
    raw buffers."""
    if t.is_cuda and _C is not None:
        _C.scale_inplace(t, float(factor))
        return t
    return t.mul_(factor)


def scale_put(dst: torch.Tensor, src: torch.Tensor, weight: float) -> None:
    """dst = weight * src, one pass. dst may be an IPC-mapped peer buffer:
    the stores traverse xGMI directly (the one-sided win_put data plane)."""
    if dst.is_cuda and _C is not None:
        _C.scale_put(dst, src, float(weight))
        return
    dst.copy_(src)
    if weight != 1.0:
        dst.mul_(weight)


def accum_put(dst: torch.Tensor, src: torch.Tensor, weight: float) -> None:
    """dst += weight * src (single-writer read-modify-write over xGMI for
    win_accumulate; no atomics needed — each origin owns its target slot)."""
    if dst.is_cuda and _C is not None:
        _C.accum_put(dst, src, float(weight))
        return
    dst.add_(src, alpha=weight)
