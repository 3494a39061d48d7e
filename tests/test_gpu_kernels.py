# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""Numerics of the native CDNA4 kernels vs plain torch fp32 references, on a
real MI355X. Single-GPU, single-process."""

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def dev():
    assert torch.cuda.is_available()
    return torch.device("cuda:0")


def _ext():
    from bluefog_amd.ops import hip_ext

    assert hip_ext.has_extension(), "bluefog_amd._C must load on a GPU box"
    return hip_ext


@pytest.mark.parametrize(
    "dtype,atol",
    [
        (torch.float32, 1e-6),
        (torch.float64, 1e-12),
        (torch.float16, 2e-3),
        (torch.bfloat16, 2e-2),
    ],
)
@pytest.mark.parametrize("numel", [1, 7, 1024, 1 << 20, (1 << 20) + 3])
def test_weighted_combine_numerics(dev, dtype, atol, numel):
    he = _ext()
    torch.manual_seed(0)
    n_nbr = 3
    self_t = torch.randn(numel, device=dev).to(dtype)
    gathered = torch.randn(n_nbr * numel, device=dev).to(dtype)
    weights = [0.1, 0.25, 0.4]
    self_w = 0.25
    out = torch.empty_like(self_t)
    he.weighted_combine(out, self_t, self_w, gathered, weights)
    # plain torch reference in the kernel's accumulation precision
    # (fp32 for f32/f16/bf16 inputs, fp64 for f64)
    acc_dtype = torch.float64 if dtype == torch.float64 else torch.float32
    acc = self_w * self_t.to(acc_dtype)
    for k in range(n_nbr):
        acc += weights[k] * gathered[k * numel : (k + 1) * numel].to(acc_dtype)
    torch.cuda.synchronize()
    assert torch.allclose(out.to(acc_dtype), acc.to(dtype).to(acc_dtype), atol=atol), (
        dtype,
        numel,
        (out.to(acc_dtype) - acc).abs().max().item(),
    )


def test_weighted_combine_many_neighbors(dev):
    """> kMaxNbrPerLaunch neighbors exercises host-side chunking."""
    he = _ext()
    torch.manual_seed(1)
    numel, n_nbr = 4096, 37
    self_t = torch.randn(numel, device=dev)
    gathered = torch.randn(n_nbr * numel, device=dev)
    weights = [1.0 / (n_nbr + 1)] * n_nbr
    out = torch.empty_like(self_t)
    he.weighted_combine(out, self_t, 1.0 / (n_nbr + 1), gathered, weights)
    ref = self_t / (n_nbr + 1) + gathered.view(n_nbr, numel).sum(0) / (n_nbr + 1)
    torch.cuda.synchronize()
    assert torch.allclose(out, ref, atol=1e-5)


def test_zero_neighbors(dev):
    he = _ext()
    t = torch.randn(1000, device=dev)
    out = torch.empty_like(t)
    he.weighted_combine(out, t, 0.5, None, [])
    torch.cuda.synchronize()
    assert torch.allclose(out, 0.5 * t)


def test_scale_put_accum(dev):
    he = _ext()
    src = torch.randn(12345, device=dev)
    dst = torch.randn(12345, device=dev)
    dst0 = dst.clone()
    he.scale_put(dst, src, 0.3)
    torch.cuda.synchronize()
    assert torch.allclose(dst, 0.3 * src, atol=1e-6)
    he.accum_put(dst, src, 0.7)
    torch.cuda.synchronize()
    assert torch.allclose(dst, src, atol=1e-5)
    he.scale(dst, 2.0)
    torch.cuda.synchronize()
    assert torch.allclose(dst, 2 * src, atol=1e-5)
    del dst0


def test_combine_sgd_matches_torch(dev):
    """Fused avg+SGD(momentum) == weighted_combine then torch.optim.SGD."""
    from bluefog_amd import _C

    torch.manual_seed(2)
    numel = 1 << 16
    p0 = torch.randn(numel, device=dev)
    grad = torch.randn(numel, device=dev)
    gathered = torch.randn(2 * numel, device=dev)
    weights = [0.3, 0.3]
    self_w = 0.4
    lr, mu, wd = 0.1, 0.9, 1e-4

    # reference: combine, then torch SGD one step with pre-seeded momentum
    ref_p = (
        self_w * p0
        + weights[0] * gathered[:numel]
        + weights[1] * gathered[numel:]
    ).clone()
    mom0 = torch.randn(numel, device=dev)
    ref_param = torch.nn.Parameter(ref_p.clone())
    opt = torch.optim.SGD([ref_param], lr=lr, momentum=mu, weight_decay=wd)
    opt.state[ref_param]["momentum_buffer"] = mom0.clone()
    ref_param.grad = grad.clone()
    opt.step()

    p = p0.clone()
    mom = mom0.clone()
    _C.weighted_combine_sgd(p, self_w, gathered, weights, grad, mom, lr, mu, wd, 0.0, False)
    torch.cuda.synchronize()
    assert torch.allclose(p, ref_param.data, atol=1e-5), (
        (p - ref_param.data).abs().max().item()
    )
    assert torch.allclose(mom, opt.state[ref_param]["momentum_buffer"], atol=1e-5)


def test_combine_adam_matches_torch(dev):
    from bluefog_amd import _C

    torch.manual_seed(3)
    numel = 1 << 14
    p0 = torch.randn(numel, device=dev)
    grad = torch.randn(numel, device=dev)
    lr, b1, b2, eps, wd = 1e-3, 0.9, 0.999, 1e-8, 0.0
    step = 1

    ref_param = torch.nn.Parameter(p0.clone())
    opt = torch.optim.Adam([ref_param], lr=lr, betas=(b1, b2), eps=eps)
    ref_param.grad = grad.clone()
    opt.step()

    p = p0.clone()
    exp_avg = torch.zeros(numel, device=dev)
    exp_avg_sq = torch.zeros(numel, device=dev)
    _C.weighted_combine_adam(
        p, 1.0, p, [], grad, exp_avg, exp_avg_sq, lr, b1, b2, eps, wd, step
    )
    torch.cuda.synchronize()
    assert torch.allclose(p, ref_param.data, atol=1e-6), (
        (p - ref_param.data).abs().max().item()
    )


def test_combine_bandwidth(dev):
    """The weighted-average stream should run HBM-bound: with 1 neighbor it
    moves 3 values per element (2 reads + 1 write). Loose floor: 2 TB/s."""
    he = _ext()
    numel = 128 * 1024 * 1024  # 512 MB per stream, fp32
    self_t = torch.randn(numel, device=dev)
    gathered = torch.randn(numel, device=dev)
    out = torch.empty_like(self_t)
    for _ in range(3):
        he.weighted_combine(out, self_t, 0.5, gathered, [0.5])
    torch.cuda.synchronize()
    import time

    t0 = time.perf_counter()
    iters = 10
    for _ in range(iters):
        he.weighted_combine(out, self_t, 0.5, gathered, [0.5])
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    tbps = 3 * numel * 4 / dt / 1e12
    print(f"\nweighted_combine effective bandwidth: {tbps:.2f} TB/s")
    assert tbps > 2.0, f"combine kernel too slow: {tbps:.2f} TB/s"


def test_gpu_neighbor_allreduce_single():
    import bluefog_amd as bf

    bf.init()
    t = torch.ones(1000, device="cuda")
    out = bf.neighbor_allreduce(t)
    torch.cuda.synchronize()
    assert torch.allclose(out, t)


def test_resnet_step_gpu():
    import bluefog_amd as bf
    from bluefog_amd.models import resnet50

    if not bf._ctx().is_initialized():
        bf.init()
    model = resnet50().cuda()
    opt = bf.DistributedAdaptWithCombineOptimizer(
        torch.optim.SGD(model.parameters(), lr=0.01, momentum=0.9),
        model=model,
        communication_type=bf.CommunicationType.neighbor_allreduce,
    )
    x = torch.randn(8, 3, 224, 224, device="cuda")
    y = torch.randint(0, 1000, (8,), device="cuda")
    lf = torch.nn.CrossEntropyLoss()
    for _ in range(3):
        opt.zero_grad()
        loss = lf(model(x), y)
        loss.backward()
        opt.step()
    torch.cuda.synchronize()
    assert torch.isfinite(loss)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16, torch.float16])
@pytest.mark.parametrize("channels_last", [False, True])
def test_add_relu_gpu_matches_eager(dev, dtype, channels_last):
    from bluefog_amd.ops.fused_modules import add_relu

    torch.manual_seed(0)
    shape = (8, 32, 14, 14)
    mf = torch.channels_last if channels_last else torch.contiguous_format
    a = torch.randn(shape, device=dev).to(dtype).to(memory_format=mf).requires_grad_()
    b = torch.randn(shape, device=dev).to(dtype).to(memory_format=mf).requires_grad_()
    a2 = a.detach().clone().requires_grad_()
    b2 = b.detach().clone().requires_grad_()
    out = add_relu(a, b)
    ref = torch.relu(a2 + b2)
    torch.cuda.synchronize()
    assert torch.equal(out, ref)
    g = torch.randn(shape, device=dev).to(dtype).to(memory_format=mf)
    out.backward(g)
    ref.backward(g)
    torch.cuda.synchronize()
    assert torch.equal(a.grad, a2.grad)
    assert torch.equal(b.grad, b2.grad)


@pytest.mark.parametrize(
    "dtype,atol,gatol",
    [
        (torch.float32, 1e-5, 2e-3),
        (torch.float16, 2e-2, 5e-2),
        (torch.bfloat16, 5e-2, 5e-2),
    ],
)
@pytest.mark.parametrize("shape", [(4, 128, 768), (2, 512, 768), (3, 7, 1000), (1, 1, 4096)])
def test_ln_add_matches_torch(dev, dtype, atol, gatol, shape):
    """Fused residual add + LayerNorm vs torch fp32 reference: forward,
    dx (both branches) and dgamma/dbeta."""
    from bluefog_amd.ops.fused_modules import _AddLayerNorm

    torch.manual_seed(12)
    H = shape[-1]
    x = torch.randn(shape, device=dev, dtype=dtype, requires_grad=True)
    r = torch.randn(shape, device=dev, dtype=dtype, requires_grad=True)
    w = torch.randn(H, device=dev, dtype=torch.float32, requires_grad=True)
    b = torch.randn(H, device=dev, dtype=torch.float32, requires_grad=True)
    y = _AddLayerNorm.apply(x, r, w, b, 1e-12)
    # fp32 torch reference
    x32 = x.detach().float().requires_grad_()
    r32 = r.detach().float().requires_grad_()
    w32 = w.detach().clone().requires_grad_()
    b32 = b.detach().clone().requires_grad_()
    y_ref = torch.nn.functional.layer_norm(x32 + r32, (H,), w32, b32, 1e-12)
    torch.cuda.synchronize()
    assert torch.allclose(y.float(), y_ref, atol=atol), (
        (y.float() - y_ref).abs().max().item()
    )
    g = torch.randn_like(y_ref)
    y.backward(g.to(dtype))
    y_ref.backward(g)
    torch.cuda.synchronize()
    for got, ref in [(x.grad, x32.grad), (r.grad, r32.grad)]:
        assert torch.allclose(got.float(), ref, atol=atol * 4), (
            (got.float() - ref).abs().max().item()
        )
    # column reductions sum `rows` input-precision-rounded products of
    # O(1) values: the error random-walks as sqrt(rows) (and so does the
    # magnitude of the sum — relative precision is preserved)
    rows = shape[0] * shape[1]
    col_atol = gatol * max(1.0, rows ** 0.5)
    assert torch.allclose(w.grad, w32.grad, atol=col_atol), (
        (w.grad - w32.grad).abs().max().item()
    )
    assert torch.allclose(b.grad, b32.grad, atol=col_atol), (
        (b.grad - b32.grad).abs().max().item()
    )


def test_fused_add_layernorm_module(dev):
    """FusedAddLayerNorm(x, residual) == nn.LayerNorm(x + residual) in
    training, including through a BertLayer-shaped composite."""
    from bluefog_amd.ops.fused_modules import FusedAddLayerNorm

    torch.manual_seed(3)
    m = FusedAddLayerNorm(768).to(dev)
    ref = torch.nn.LayerNorm(768).to(dev)
    ref.load_state_dict(m.state_dict())
    x = torch.randn(4, 64, 768, device=dev)
    r = torch.randn(4, 64, 768, device=dev)
    out = m(x, r)
    expected = ref(x + r)
    torch.cuda.synchronize()
    assert torch.allclose(out, expected, atol=1e-5), (
        (out - expected).abs().max().item()
    )
    # no-residual call degrades to plain LayerNorm
    assert torch.allclose(m(x), ref(x), atol=1e-6)
