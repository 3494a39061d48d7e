# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""Numerics of the fused AWC average+step path on GPU: training with the
fused bucket kernels must match plain torch training step-for-step."""

import copy

import pytest
import torch
import torch.nn as nn

pytestmark = pytest.mark.gpu


def _models():
    torch.manual_seed(7)
    m1 = nn.Sequential(
        nn.Linear(64, 128), nn.ReLU(), nn.Linear(128, 64), nn.ReLU(), nn.Linear(64, 10)
    ).cuda()
    m2 = copy.deepcopy(m1)
    return m1, m2


@pytest.mark.parametrize("momentum", [0.0, 0.9])
def test_fused_sgd_matches_plain_training(momentum):
    import bluefog_amd as bf

    if not bf._ctx().is_initialized():
        bf.init()
    m_ref, m_fused = _models()
    opt_ref = torch.optim.SGD(m_ref.parameters(), lr=0.05, momentum=momentum,
                              weight_decay=1e-4)
    opt_fused = bf.DistributedAdaptWithCombineOptimizer(
        torch.optim.SGD(m_fused.parameters(), lr=0.05, momentum=momentum,
                        weight_decay=1e-4),
        model=m_fused,
        communication_type=bf.CommunicationType.neighbor_allreduce,
    )
    assert opt_fused._fused == "sgd", "fused SGD mode must engage on GPU"
    torch.manual_seed(11)
    xs = [torch.randn(32, 64, device="cuda") for _ in range(5)]
    ys = [torch.randint(0, 10, (32,), device="cuda") for _ in range(5)]
    lf = nn.CrossEntropyLoss()
    for x, y in zip(xs, ys):
        opt_ref.zero_grad()
        lf(m_ref(x), y).backward()
        opt_ref.step()
        opt_fused.zero_grad()
        lf(m_fused(x), y).backward()
        opt_fused.step()
    torch.cuda.synchronize()
    for p_ref, p_f in zip(m_ref.parameters(), m_fused.parameters()):
        assert torch.allclose(p_ref, p_f, atol=1e-5), (
            (p_ref - p_f).abs().max().item()
        )


def test_fused_adam_matches_plain_training():
    import bluefog_amd as bf

    if not bf._ctx().is_initialized():
        bf.init()
    m_ref, m_fused = _models()
    opt_ref = torch.optim.Adam(m_ref.parameters(), lr=1e-3)
    opt_fused = bf.DistributedAdaptWithCombineOptimizer(
        torch.optim.Adam(m_fused.parameters(), lr=1e-3),
        model=m_fused,
        communication_type=bf.CommunicationType.neighbor_allreduce,
    )
    assert opt_fused._fused == "adam"
    torch.manual_seed(12)
    lf = nn.CrossEntropyLoss()
    for _ in range(5):
        x = torch.randn(32, 64, device="cuda")
        y = torch.randint(0, 10, (32,), device="cuda")
        opt_ref.zero_grad()
        lf(m_ref(x), y).backward()
        opt_ref.step()
        opt_fused.zero_grad()
        lf(m_fused(x), y).backward()
        opt_fused.step()
    torch.cuda.synchronize()
    for p_ref, p_f in zip(m_ref.parameters(), m_fused.parameters()):
        assert torch.allclose(p_ref, p_f, atol=1e-5), (
            (p_ref - p_f).abs().max().item()
        )


def test_fused_atc_unused_head_still_steps():
    """A param whose grad hook never fires (unused head) keeps its bucket's
    pending set non-empty; synchronize() must still step+exchange the fired
    params of that bucket instead of silently skipping them."""
    import bluefog_amd as bf

    if not bf._ctx().is_initialized():
        bf.init()

    class M(nn.Module):
        def __init__(self):
            super().__init__()
            torch.manual_seed(13)
            self.used = nn.Linear(16, 16, bias=False)
            self.unused = nn.Linear(16, 16, bias=False)  # not in forward

        def forward(self, x):
            return self.used(x)

    m_fused = M().cuda()
    m_ref = copy.deepcopy(m_fused)
    opt = bf.DistributedAdaptThenCombineOptimizer(
        torch.optim.SGD(m_fused.parameters(), lr=0.1),
        model=m_fused,
        communication_type=bf.CommunicationType.neighbor_allreduce,
    )
    assert opt._fused == "sgd"
    x = torch.randn(8, 16, device="cuda")
    opt.zero_grad()
    m_fused(x).sum().backward()
    opt.step()
    torch.cuda.synchronize()
    # reference step on the used layer only
    m_ref(x).sum().backward()
    with torch.no_grad():
        m_ref.used.weight -= 0.1 * m_ref.used.weight.grad
    assert torch.allclose(m_fused.used.weight, m_ref.used.weight, atol=1e-5), (
        (m_fused.used.weight - m_ref.used.weight).abs().max().item()
    )
    assert torch.equal(m_fused.unused.weight, m_ref.unused.weight)


@pytest.mark.parametrize("kind", ["sgd", "adam"])
def test_fused_atc_matches_plain_training(kind):
    """ATC at world size 1: the fused bucket kernels apply the optimizer
    update from the backward hooks; the result must equal plain torch
    training step-for-step."""
    import bluefog_amd as bf

    if not bf._ctx().is_initialized():
        bf.init()
    m_ref, m_fused = _models()
    if kind == "sgd":
        opt_ref = torch.optim.SGD(m_ref.parameters(), lr=0.05, momentum=0.9,
                                  weight_decay=1e-4)
        base = torch.optim.SGD(m_fused.parameters(), lr=0.05, momentum=0.9,
                               weight_decay=1e-4)
    else:
        opt_ref = torch.optim.Adam(m_ref.parameters(), lr=1e-3)
        base = torch.optim.Adam(m_fused.parameters(), lr=1e-3)
    opt_fused = bf.DistributedAdaptThenCombineOptimizer(
        base, model=m_fused,
        communication_type=bf.CommunicationType.neighbor_allreduce,
    )
    assert opt_fused._fused == kind, "fused ATC mode must engage on GPU"
    torch.manual_seed(21)
    lf = torch.nn.CrossEntropyLoss()
    for _ in range(5):
        x = torch.randn(32, 64, device="cuda")
        y = torch.randint(0, 10, (32,), device="cuda")
        opt_ref.zero_grad()
        lf(m_ref(x), y).backward()
        opt_ref.step()
        opt_fused.zero_grad()
        lf(m_fused(x), y).backward()
        opt_fused.step()
    torch.cuda.synchronize()
    for p_ref, p_f in zip(m_ref.parameters(), m_fused.parameters()):
        assert torch.allclose(p_ref, p_f, atol=1e-5), (
            (p_ref - p_f).abs().max().item()
        )
