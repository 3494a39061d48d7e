# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""Fused elementwise modules for model hot paths.

``add_relu(a, b)`` computes ``relu(a + b)`` — the residual join of every
ResNet block — as ONE gfx950 kernel forward (torch runs an add kernel plus
a clamp kernel: one extra full read+write pass over the activation tensor)
and one masked pass backward that produces the shared gradient for both
branches. Falls back to eager torch on CPU, for mismatched layouts, or
when the extension is absent (CPU tests compare the two paths).
"""

import torch

from bluefog_amd.ops import hip_ext


def _dense(t: torch.Tensor) -> bool:
    return t.is_contiguous() or t.is_contiguous(memory_format=torch.channels_last)


class _AddReLU(torch.autograd.Function):
    @staticmethod
    def forward(ctx, a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
        from bluefog_amd import _C

        out = torch.empty_like(a)
        _C.add_relu_fwd(out, a, b)
        ctx.save_for_backward(out)
        return out

    @staticmethod
    def backward(ctx, grad_out: torch.Tensor):
        from bluefog_amd import _C

        (out,) = ctx.saved_tensors
        if grad_out.stride() != out.stride():
            grad_out = grad_out.contiguous(
                memory_format=torch.channels_last
                if out.is_contiguous(memory_format=torch.channels_last)
                and out.dim() == 4
                else torch.contiguous_format
            )
        gin = torch.empty_like(grad_out)
        _C.relu_bwd_mask(gin, grad_out, out)
        # both branches of the residual receive the same gradient
        return gin, gin


def add_relu(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """relu(a + b) — fused on gfx950, eager elsewhere."""
    if (
        a.is_cuda
        and hip_ext.has_extension()
        and a.dtype == b.dtype
        and a.shape == b.shape
        and a.stride() == b.stride()
        and _dense(a)
        and _dense(b)
        and a.dtype in (torch.float32, torch.float64, torch.float16, torch.bfloat16)
    ):
        return _AddReLU.apply(a, b)
    return torch.relu(a + b)
