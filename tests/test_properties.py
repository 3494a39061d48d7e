# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""Property-based tests (hypothesis): topology invariants that the
decentralized algorithms depend on, and the exact math of the op fallbacks
used as the CPU reference for the GPU kernels."""

import numpy as np
import torch
from hypothesis import given, settings
from hypothesis import strategies as st

import bluefog_amd.parallel.topology as tu
from bluefog_amd.ops import hip_ext


# ---------------------------------------------------------------------------
# topology invariants
# ---------------------------------------------------------------------------

_FAMILIES = [
    ("exp2", lambda n: tu.ExponentialTwoGraph(n), lambda n: n >= 2),
    ("exp3", lambda n: tu.ExponentialGraph(n, base=3), lambda n: n >= 2),
    ("ring", lambda n: tu.RingGraph(n), lambda n: n >= 2),
    ("mesh", lambda n: tu.MeshGrid2DGraph(n), lambda n: n >= 2),
    ("star", lambda n: tu.StarGraph(n), lambda n: n >= 2),
    ("full", lambda n: tu.FullyConnectedGraph(n), lambda n: n >= 2),
    ("symexp4", lambda n: tu.SymmetricExponentialGraph(n, base=4), lambda n: n >= 2),
]


@settings(max_examples=40, deadline=None)
@given(
    n=st.integers(min_value=2, max_value=33),
    fam=st.sampled_from(range(len(_FAMILIES))),
)
def test_recv_weights_row_stochastic(n, fam):
    """GetRecvWeights must return a convex combination (weights sum to 1,
    all >= 0) for every graph family and size — the property neighbor
    averaging needs to preserve the global mean."""
    name, build, ok = _FAMILIES[fam]
    if not ok(n):
        return
    topo = build(n)
    for rank in range(min(n, 8)):
        self_w, nbr_w = tu.GetRecvWeights(topo, rank)
        total = self_w + sum(nbr_w.values())
        assert abs(total - 1.0) < 1e-9, (name, n, rank, total)
        assert self_w >= 0 and all(w >= -1e-12 for w in nbr_w.values())
        assert set(nbr_w) == set(
            int(r) for r in topo.predecessors(rank) if r != rank
        ), (name, n, rank)


@settings(max_examples=30, deadline=None)
@given(n=st.integers(min_value=2, max_value=32), seed=st.integers(0, 10**6))
def test_dynamic_one_peer_sets_are_mutually_consistent(n, seed):
    """Across all ranks at the same iteration index, send/recv sets of
    GetDynamicOnePeerSendRecvRanks must mirror each other."""
    if n & (n - 1):
        n = 1 << (n.bit_length() - 1)  # power of two for exp2
    if n < 2:
        return
    topo = tu.ExponentialTwoGraph(n)
    gens = [tu.GetDynamicOnePeerSendRecvRanks(topo, r) for r in range(n)]
    for _ in range(3):
        sends, recvs = zip(*(next(g) for g in gens))
        for r in range(n):
            for dst in sends[r]:
                assert r in recvs[dst], (n, r, dst)
            for src in recvs[r]:
                assert r in sends[src], (n, r, src)


@settings(max_examples=25, deadline=None)
@given(n=st.integers(2, 64))
def test_topology_equivalence_reflexive(n):
    topo = tu.RingGraph(n)
    assert tu.IsTopologyEquivalent(topo, tu.RingGraph(n))
    assert not tu.IsTopologyEquivalent(topo, tu.FullyConnectedGraph(n)) or n <= 3


# ---------------------------------------------------------------------------
# op math fallbacks (CPU reference used to validate the HIP kernels)
# ---------------------------------------------------------------------------


@settings(max_examples=40, deadline=None)
@given(
    numel=st.integers(1, 300),
    n_nbr=st.integers(0, 5),
    seed=st.integers(0, 10**6),
)
def test_weighted_combine_fallback_matches_numpy(numel, n_nbr, seed):
    rng = np.random.default_rng(seed)
    self_t = torch.tensor(rng.standard_normal(numel), dtype=torch.float64)
    gathered = torch.tensor(rng.standard_normal(n_nbr * numel), dtype=torch.float64)
    weights = rng.uniform(0, 1, n_nbr).tolist()
    self_w = float(rng.uniform(0, 1))
    out = torch.empty_like(self_t)
    hip_ext.weighted_combine(out, self_t, self_w, gathered if n_nbr else None, weights)
    ref = self_w * self_t.numpy().copy()
    for k in range(n_nbr):
        ref = ref + weights[k] * gathered.numpy()[k * numel : (k + 1) * numel]
    np.testing.assert_allclose(out.numpy(), ref, atol=1e-12)


@settings(max_examples=30, deadline=None)
@given(
    numel=st.integers(1, 200),
    momentum=st.sampled_from([0.0, 0.9]),
    nesterov=st.booleans(),
    seed=st.integers(0, 10**6),
)
def test_combine_sgd_fallback_matches_torch_sgd(numel, momentum, nesterov, seed):
    if nesterov and momentum == 0.0:
        return
    rng = np.random.default_rng(seed)
    p1 = torch.tensor(rng.standard_normal(numel), dtype=torch.float64, requires_grad=True)
    p2 = p1.detach().clone().requires_grad_()
    grad = torch.tensor(rng.standard_normal(numel), dtype=torch.float64)
    opt = torch.optim.SGD([p2], lr=0.05, momentum=momentum, weight_decay=1e-3,
                          nesterov=nesterov)
    mom_buf = torch.zeros_like(p1) if momentum != 0 else None
    for _ in range(3):
        with torch.no_grad():
            hip_ext.weighted_combine_sgd(
                p1.data, 1.0, None, [], grad, mom_buf, 0.05, momentum, 1e-3,
                0.0, nesterov,
            )
        opt.zero_grad()
        p2.grad = grad.clone()
        opt.step()
    # torch SGD's first momentum step uses buf=grad (no (1-dampening) scale);
    # our kernel matches that convention
    np.testing.assert_allclose(p1.detach().numpy(), p2.detach().numpy(), atol=1e-10)


@settings(max_examples=20, deadline=None)
@given(
    machines=st.sampled_from([2, 4, 8]),
    local=st.sampled_from([2, 4]),
    rounds=st.integers(1, 4),
)
def test_machine_generator_mutually_consistent(machines, local, rounds):
    """GetExp2DynamicSendRecvMachineRanks: at every iteration the machine-
    level send/recv sets must mirror across the leader ranks."""
    world = machines * local
    gens = {
        r: tu.GetExp2DynamicSendRecvMachineRanks(
            world_size=world, local_size=local, self_rank=r, local_rank=r % local
        )
        for r in range(world)
    }
    for _ in range(rounds):
        sends, recvs = {}, {}
        for r, g in gens.items():
            s_m, r_m = next(g)
            m = r // local
            sends[m] = set(s_m)
            recvs[m] = set(r_m)
        for m, dsts in sends.items():
            for d in dsts:
                assert m in recvs[d], (machines, local, m, d, sends, recvs)
            for srcm in recvs[m]:
                assert m in sends[srcm]
