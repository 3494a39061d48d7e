# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""add_relu fused-module tests. CPU path = eager fallback; the GPU kernel
is validated against the same eager math in tests/test_gpu_kernels.py."""

import torch

from bluefog_amd.ops.fused_modules import add_relu


def test_add_relu_cpu_matches_eager():
    torch.manual_seed(0)
    a = torch.randn(64, 8, requires_grad=True)
    b = torch.randn(64, 8, requires_grad=True)
    out = add_relu(a, b)
    ref = torch.relu(a + b)
    assert torch.equal(out, ref)
    g = torch.randn_like(out)
    out.backward(g)
    ga, gb = a.grad.clone(), b.grad.clone()
    a.grad = b.grad = None
    ref.backward(g)
    assert torch.equal(ga, a.grad)
    assert torch.equal(gb, b.grad)


def test_resnet_forward_backward_cpu():
    from bluefog_amd.models import resnet18

    m = resnet18(num_classes=10)
    x = torch.randn(2, 3, 32, 32)
    y = m(x).sum()
    y.backward()
    assert all(p.grad is not None for p in m.parameters())
