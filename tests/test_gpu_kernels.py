"""GPU numerics tests: every HIP kernel vs the plain-torch fp32 reference
(murmura_amd/ops/reference.py) on the same inputs."""

import pytest
import torch
from torch import nn

pytestmark = pytest.mark.gpu

from murmura_amd import ops
from murmura_amd.ops import reference as ref


@pytest.fixture(scope="module", autouse=True)
def _require_native():
    assert ops.native_available(), "HIP extension must be present on the GPU box"


def _rand(shape, dtype=torch.float32, seed=0):
    g = torch.Generator().manual_seed(seed)
    return torch.randn(*shape, generator=g).to("cuda", dtype)


P_ODD = 1_000_003  # odd size exercises the scalar tail


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("m,p", [(2, 1024), (8, P_ODD), (16, 65537)])
def test_weighted_sum(dtype, m, p):
    x = _rand((m, p), dtype)
    w = torch.rand(m, device="cuda")
    out = ops.weighted_sum(x, w)
    expect = ref.weighted_sum(x.float().cpu(), w.cpu())
    tol = 1e-4 if dtype == torch.float32 else 0.05
    assert out.dtype == dtype
    assert torch.allclose(out.float().cpu(), expect, atol=tol, rtol=tol)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("m", [2, 5, 8, 16])
def test_pairwise_sq_dists(dtype, m):
    x = _rand((m, P_ODD), dtype, seed=m)
    d2 = ops.pairwise_sq_dists(x)
    expect = ref.pairwise_sq_dists(x.float().cpu())
    rel = (d2.cpu() - expect).abs().max() / expect.max().clamp_min(1e-6)
    assert rel.item() < (1e-4 if dtype == torch.float32 else 2e-2)
    assert torch.all(d2.diagonal().abs() / expect.max().clamp_min(1e-6) < 1e-4)


@pytest.mark.parametrize("m", [11, 13, 20, 33, 64, 65])
def test_pairwise_tiled_and_fallback(m):
    x = _rand((m, 10_001), seed=m)
    d2 = ops.pairwise_sq_dists(x)
    expect = ref.pairwise_sq_dists(x.float().cpu())
    assert torch.allclose(d2.cpu(), expect, rtol=1e-3, atol=1e-2)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_row_norms_and_dists(dtype):
    x = _rand((6, P_ODD), dtype, seed=3)
    own = _rand((P_ODD,), dtype, seed=4)
    n = ops.row_norms(x)
    expect_n = ref.row_norms(x.float().cpu())
    assert torch.allclose(n.cpu(), expect_n, rtol=1e-3)
    d = ops.l2_dists_to(own, x)
    expect_d = ref.l2_dists_to(own.float().cpu(), x.float().cpu())
    assert torch.allclose(d.cpu(), expect_d, rtol=1e-3)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("sketch_size", [1000, 8192])
def test_count_sketch(dtype, sketch_size):
    m, p = 3, 500_009
    x = _rand((m, p), dtype, seed=5)
    h, s = ref.make_sketch_tables(p, sketch_size, seed=7, device=torch.device("cuda"))
    out = ops.count_sketch(x, h, s, sketch_size)
    expect = ref.count_sketch(x.float().cpu(), h.cpu(), s.cpu(), sketch_size)
    scale = expect.abs().max().clamp_min(1e-6)
    assert ((out.cpu() - expect).abs().max() / scale).item() < (
        1e-4 if dtype == torch.float32 else 2e-2
    )


def test_count_sketch_large_s_fallback():
    """m*S too large for replicated LDS histograms: the per-row fallback
    kernel must produce the same sketch."""
    m, p, S = 2, 200_003, 20_000
    x = _rand((m, p), torch.float32, seed=11)
    h, s = ref.make_sketch_tables(p, S, seed=13, device=torch.device("cuda"))
    out = ops.count_sketch(x, h, s, S)
    expect = ref.count_sketch(x.float().cpu(), h.cpu(), s.cpu(), S)
    scale = expect.abs().max().clamp_min(1e-6)
    assert ((out.cpu() - expect).abs().max() / scale).item() < 1e-4


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_sgd_step(dtype):
    p = _rand((P_ODD,), dtype, seed=8)
    g = _rand((P_ODD,), dtype, seed=9)
    expect = (p.float() - 0.05 * g.float()).to(dtype)
    ops.sgd_step(p, g, 0.05)
    assert torch.allclose(p.float(), expect.float(), atol=1e-6)


def test_gaussian_inject_statistics_and_determinism():
    x = torch.zeros(2_000_003, device="cuda")
    a = ops.gaussian_inject(x, 3.0, seed=11, offset=5)
    b = ops.gaussian_inject(x, 3.0, seed=11, offset=5)
    c = ops.gaussian_inject(x, 3.0, seed=11, offset=6)
    assert torch.equal(a, b)
    assert not torch.equal(a, c)
    assert abs(a.mean().item()) < 0.01
    assert abs(a.std().item() - 3.0) < 0.01
    # normality sanity: ~68.3% within 1 sigma
    frac = (a.abs() < 3.0).float().mean().item()
    assert abs(frac - 0.683) < 0.01


def test_gaussian_inject_bf16():
    x = torch.zeros(1_000_000, device="cuda", dtype=torch.bfloat16)
    a = ops.gaussian_inject(x, 2.0, seed=1, offset=0)
    assert a.dtype == torch.bfloat16
    assert abs(a.float().std().item() - 2.0) < 0.05


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_scale_inject(dtype):
    x = _rand((P_ODD,), dtype)
    out = ops.scale_inject(x, -5.0)
    assert torch.allclose(out.float(), -5.0 * x.float(), rtol=1e-2, atol=1e-3)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("b,c", [(64, 10), (333, 62), (128, 7)])
def test_ce_loss_acc(dtype, b, c):
    logits = _rand((b, c), dtype, seed=b)
    targets = torch.randint(0, c, (b,), device="cuda")
    loss, correct = ops.ce_loss_acc(logits, targets)
    eloss, ecorrect = ref.ce_loss_acc(logits.float().cpu(), targets.cpu())
    assert correct.item() == ecorrect.item()
    assert abs(loss.item() - eloss.item()) / max(1.0, eloss.item()) < (
        1e-4 if dtype == torch.float32 else 1e-2
    )


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("b,c", [(64, 6), (100, 12), (222, 100)])
def test_evidential_stats(dtype, b, c):
    logits = _rand((b, c), dtype, seed=b + c)
    targets = torch.randint(0, c, (b,), device="cuda")
    v, e, s, corr = ops.evidential_stats(logits, targets)
    ev, ee, es, ec = ref.evidential_stats(logits.float().cpu(), targets.cpu())
    tol = 1e-3 if dtype == torch.float32 else 2e-2
    assert abs(v.item() - ev.item()) / max(1.0, ev.item()) < tol
    assert abs(e.item() - ee.item()) / max(1.0, ee.item()) < tol
    assert abs(s.item() - es.item()) / max(1.0, es.item()) < tol
    assert corr.item() == ec.item()


def test_gpu_aggregators_end_to_end():
    """All six aggregators on GPU flat states produce finite outputs matching
    the CPU torch path."""
    import os

    from murmura_amd.aggregation import (
        BALANCEAggregator,
        FedAvgAggregator,
        KrumAggregator,
        SketchguardAggregator,
    )

    P = 100_003
    own = _rand((P,))
    nbrs = _rand((5, P), seed=2)
    for agg_cls, kw in [
        (FedAvgAggregator, {}),
        (KrumAggregator, {"num_compromised": 1}),
        (BALANCEAggregator, {}),
        (SketchguardAggregator, {"model_dim": P}),
    ]:
        out_gpu = agg_cls(**kw).aggregate(0, own, nbrs, round_num=1)
        out_cpu = agg_cls(**kw).aggregate(0, own.cpu(), nbrs.cpu(), round_num=1)
        assert torch.isfinite(out_gpu).all()
        assert torch.allclose(out_gpu.cpu(), out_cpu, atol=1e-3, rtol=1e-3), agg_cls.__name__


def test_native_required_on_gpu(monkeypatch):
    """A cuda tensor with the extension 'missing' must raise, never silently
    fall back to eager."""
    import murmura_amd.ops as O

    monkeypatch.setattr(O, "_EXT", None)
    monkeypatch.setattr(O, "_EXT_ERR", "simulated-missing")
    x = torch.randn(10, device="cuda")
    with pytest.raises(RuntimeError, match="HIP extension"):
        O.weighted_sum(x.view(1, -1).contiguous(), torch.ones(1, device="cuda"))


# ---------------------------------------------------------------- K14 fused BN
@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("c", [64, 128, 256, 512])
def test_fused_bn_forward_backward_matches_torch(dtype, c):
    from murmura_amd.ops.fused_bn import MurmuraBatchNorm2d

    torch.manual_seed(c)
    n, h, w = 16, 8, 8
    x = torch.randn(n, c, h, w, device="cuda", dtype=dtype)
    xf = x.contiguous(memory_format=torch.channels_last).requires_grad_(True)
    xr = x.clone().float().requires_grad_(True)

    fused = MurmuraBatchNorm2d(c).cuda().to(dtype).to(memory_format=torch.channels_last)
    refbn = nn.BatchNorm2d(c).cuda().float()
    with torch.no_grad():
        refbn.weight.copy_(fused.weight.float())
        refbn.bias.copy_(fused.bias.float())

    fused.train()
    refbn.train()
    y = fused(xf)
    yr = refbn(xr)
    tol = 2e-5 if dtype == torch.float32 else 3e-2
    assert torch.allclose(y.float(), yr, atol=tol, rtol=tol)
    # running stats updated identically (unbiased var, momentum 0.1)
    assert torch.allclose(fused.running_mean.float(), refbn.running_mean, atol=tol, rtol=1e-2)
    assert torch.allclose(fused.running_var.float(), refbn.running_var, atol=tol, rtol=1e-2)
    assert fused.num_batches_tracked.item() == 1

    g = torch.randn_like(yr)
    y.backward(g.to(dtype).contiguous(memory_format=torch.channels_last))
    yr.backward(g)
    gtol = 1e-4 if dtype == torch.float32 else 6e-2
    scale = xr.grad.abs().max().clamp_min(1e-6)
    assert ((xf.grad.float() - xr.grad).abs().max() / scale).item() < gtol
    wscale = refbn.weight.grad.abs().max().clamp_min(1e-6)
    assert ((fused.weight.grad.float() - refbn.weight.grad).abs().max() / wscale).item() < gtol
    assert ((fused.bias.grad.float() - refbn.bias.grad).abs().max() / wscale).item() < gtol


def test_fused_bn_eval_matches_torch():
    from murmura_amd.ops.fused_bn import MurmuraBatchNorm2d

    torch.manual_seed(1)
    c = 64
    fused = MurmuraBatchNorm2d(c).cuda().to(memory_format=torch.channels_last)
    with torch.no_grad():
        fused.running_mean.uniform_(-1, 1)
        fused.running_var.uniform_(0.5, 2.0)
    refbn = nn.BatchNorm2d(c).cuda()
    refbn.load_state_dict(fused.state_dict())
    fused.eval()
    refbn.eval()
    x = torch.randn(8, c, 4, 4, device="cuda")
    y = fused(x.contiguous(memory_format=torch.channels_last))
    yr = refbn(x)
    assert torch.allclose(y, yr, atol=1e-5, rtol=1e-5)


def test_fused_bn_cpu_fallback_identical_to_torch():
    from murmura_amd.ops.fused_bn import MurmuraBatchNorm2d

    m = MurmuraBatchNorm2d(16)
    r = nn.BatchNorm2d(16)
    r.load_state_dict(m.state_dict())
    x = torch.randn(4, 16, 5, 5)
    assert torch.allclose(m(x), r(x))


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_fused_bn_relu_matches_torch(dtype):
    from murmura_amd.ops.fused_bn import MurmuraBNReLU

    torch.manual_seed(5)
    c = 128
    x = torch.randn(8, c, 8, 8, device="cuda", dtype=dtype)
    xf = x.contiguous(memory_format=torch.channels_last).requires_grad_(True)
    xr = x.clone().float().requires_grad_(True)
    fused = MurmuraBNReLU(c).cuda().to(dtype)
    refbn = nn.BatchNorm2d(c).cuda().float()
    refbn.load_state_dict({k: v.float() for k, v in fused.state_dict().items()})
    fused.train(); refbn.train()
    y = fused(xf)
    yr = torch.relu(refbn(xr))
    tol = 1e-5 if dtype == torch.float32 else 3e-2
    assert torch.all(y >= 0)
    assert torch.allclose(y.float(), yr, atol=tol, rtol=tol)
    g = torch.randn_like(yr)
    y.backward(g.to(dtype).contiguous(memory_format=torch.channels_last))
    yr.backward(g)
    gtol = 1e-4 if dtype == torch.float32 else 6e-2
    scale = xr.grad.abs().max().clamp_min(1e-6)
    assert ((xf.grad.float() - xr.grad).abs().max() / scale).item() < gtol
    # eval-mode fused relu
    fused.eval(); refbn.eval()
    with torch.no_grad():
        ye = fused(x.contiguous(memory_format=torch.channels_last))
        yre = torch.relu(refbn(x.float()))
    assert torch.allclose(ye.float(), yre, atol=tol, rtol=tol)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("c", [64, 256])
def test_fused_bn_add_relu_matches_torch(dtype, c):
    """Residual-fused tail: y = relu(bn(x) + res) forward AND both input
    gradients from one backward launch (the ResNet BasicBlock tail)."""
    from murmura_amd.ops.fused_bn import MurmuraBNAddReLU

    torch.manual_seed(c)
    n, h, w = 8, 8, 8
    x = torch.randn(n, c, h, w, device="cuda", dtype=dtype)
    r = torch.randn(n, c, h, w, device="cuda", dtype=dtype)
    xf = x.contiguous(memory_format=torch.channels_last).requires_grad_(True)
    rf = r.contiguous(memory_format=torch.channels_last).requires_grad_(True)
    xr = x.clone().float().requires_grad_(True)
    rr = r.clone().float().requires_grad_(True)

    fused = MurmuraBNAddReLU(c).cuda().to(dtype)
    refbn = nn.BatchNorm2d(c).cuda().float()
    refbn.load_state_dict({k: v.float() for k, v in fused.state_dict().items()})
    fused.train(); refbn.train()

    y = fused(xf, res=rf)
    yr = torch.relu(refbn(xr) + rr)
    tol = 1e-5 if dtype == torch.float32 else 3e-2
    assert torch.allclose(y.float(), yr, atol=tol, rtol=tol)

    g = torch.randn_like(yr)
    y.backward(g.to(dtype).contiguous(memory_format=torch.channels_last))
    yr.backward(g)
    gtol = 1e-4 if dtype == torch.float32 else 6e-2
    for got, ref in [(xf.grad, xr.grad), (rf.grad, rr.grad)]:
        scale = ref.abs().max().clamp_min(1e-6)
        assert ((got.float() - ref).abs().max() / scale).item() < gtol
    wscale = refbn.weight.grad.abs().max().clamp_min(1e-6)
    assert ((fused.weight.grad.float() - refbn.weight.grad).abs().max() / wscale).item() < gtol

    # eval mode with residual
    fused.eval(); refbn.eval()
    with torch.no_grad():
        ye = fused(x.contiguous(memory_format=torch.channels_last),
                   res=r.contiguous(memory_format=torch.channels_last))
        yre = torch.relu(refbn(x.float()) + r.float())
    assert torch.allclose(ye.float(), yre, atol=tol, rtol=tol)


def test_fused_bn_add_relu_cpu_fallback():
    from murmura_amd.ops.fused_bn import MurmuraBNAddReLU

    m = MurmuraBNAddReLU(16)
    r = nn.BatchNorm2d(16)
    r.load_state_dict(m.state_dict())
    x = torch.randn(4, 16, 5, 5)
    res = torch.randn(4, 16, 5, 5)
    assert torch.allclose(m(x, res=res), torch.relu(r(x) + res))


def test_deferred_num_batches_tracked_bump():
    from murmura_amd.ops.fused_bn import (
        MurmuraBatchNorm2d,
        bump_num_batches_tracked,
    )

    m = nn.Sequential(MurmuraBatchNorm2d(8), MurmuraBatchNorm2d(8))
    bump_num_batches_tracked(m, 5)
    assert m[0].num_batches_tracked.item() == 5
    assert m[1].num_batches_tracked.item() == 5


# ---------------------------------------------------------------- K15 wrw conv
@pytest.mark.parametrize("shape", [
    (8, 64, 32, 32, 64),    # flagship layer-1 shape (smaller batch)
    (8, 128, 16, 16, 128),
    (8, 256, 8, 8, 256),
    (8, 512, 4, 4, 512),
    (4, 64, 8, 8, 128),     # C != K
])
def test_conv3x3_wrw_matches_torch(shape):
    """Hand-written MFMA weight gradient vs torch/MIOpen wgrad (fp32 ref)."""
    from murmura_amd.ops import _load_ext

    n, c, h, w, k = shape
    torch.manual_seed(c + k)
    x = torch.randn(n, c, h, w, device="cuda", dtype=torch.bfloat16)
    x = x.contiguous(memory_format=torch.channels_last)
    dy = torch.randn(n, k, h, w, device="cuda", dtype=torch.bfloat16)
    dy = dy.contiguous(memory_format=torch.channels_last)

    dw = _load_ext().conv3x3s1_wrw(x, dy)
    assert dw.shape == (k, c, 3, 3)

    ref = torch.nn.grad.conv2d_weight(
        x.float(), (k, c, 3, 3), dy.float(), stride=1, padding=1
    )
    scale = ref.abs().max().clamp_min(1e-6)
    err = (dw.float() - ref).abs().max() / scale
    assert err.item() < 5e-2, f"rel err {err.item()}"


def test_conv3x3_module_end_to_end(monkeypatch):
    """MurmuraConv3x3 inside autograd: dw and dx both correct."""
    monkeypatch.setenv("MURMURA_NATIVE_WRW", "1")
    from murmura_amd.ops.fused_conv import MurmuraConv3x3

    torch.manual_seed(0)
    m = MurmuraConv3x3(64, 64).cuda().to(torch.bfloat16)
    m = m.to(memory_format=torch.channels_last)
    x = torch.randn(4, 64, 16, 16, device="cuda", dtype=torch.bfloat16)
    x = x.contiguous(memory_format=torch.channels_last).requires_grad_(True)
    y = m(x)
    g = torch.randn_like(y)
    y.backward(g)

    xr = x.detach().float().requires_grad_(True)
    mr = nn.Conv2d(64, 64, 3, padding=1, bias=False).cuda().float()
    with torch.no_grad():
        mr.weight.copy_(m.weight.float())
    yr = mr(xr)
    yr.backward(g.float())
    ws = mr.weight.grad.abs().max().clamp_min(1e-6)
    assert ((m.weight.grad.float() - mr.weight.grad).abs().max() / ws).item() < 5e-2
    xs = xr.grad.abs().max().clamp_min(1e-6)
    assert ((x.grad.float() - xr.grad).abs().max() / xs).item() < 5e-2
