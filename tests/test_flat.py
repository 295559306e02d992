import torch
from torch import nn

from murmura_amd.core.flat import (
    FlatParamSpec,
    FlatParamStore,
    calculate_model_dimension,
    flatten_state_dict,
)
from murmura_amd.models import ResNet18, SimpleMLP, count_params


def _bn_model():
    return nn.Sequential(nn.Linear(4, 8), nn.BatchNorm1d(8), nn.ReLU(), nn.Linear(8, 3))


def test_spec_covers_params_and_float_buffers():
    m = _bn_model()
    spec = FlatParamSpec.from_model(m)
    n_params = sum(p.numel() for p in m.parameters())
    # BN adds running_mean + running_var (float buffers), excludes num_batches_tracked
    n_fbuf = sum(b.numel() for b in m.buffers() if torch.is_floating_point(b))
    assert spec.param_numel == n_params
    assert spec.total_numel == n_params + n_fbuf
    names = [e.name for e in spec]
    assert not any("num_batches_tracked" in n for n in names)


def test_store_views_are_bound():
    m = _bn_model()
    store = FlatParamStore(m, torch.device("cpu"), torch.float32)
    # mutating flat must be visible through the module's parameters
    with torch.no_grad():
        store.flat.fill_(0.5)
    for p in store.model.parameters():
        assert torch.all(p == 0.5)
    for b in store.model.buffers():
        if torch.is_floating_point(b):
            assert torch.all(b == 0.5)


def test_training_updates_flat_in_place():
    store = FlatParamStore(SimpleMLP(8, 16, 3), torch.device("cpu"), torch.float32)
    g = store.ensure_grads()
    x = torch.randn(16, 8)
    y = torch.randint(0, 3, (16,))
    before = store.flat.clone()
    store.zero_grad()
    loss = nn.functional.cross_entropy(store.model(x), y)
    loss.backward()
    assert g.abs().sum() > 0
    from murmura_amd import ops

    ops.sgd_step(store.flat[: store.spec.param_numel], g, lr=0.1)
    assert not torch.equal(before, store.flat)
    # manual check: flat == before - lr * grad on the param prefix
    pn = store.spec.param_numel
    assert torch.allclose(store.flat[:pn], before[:pn] - 0.1 * g, atol=1e-6)


def test_state_dict_roundtrip():
    m1 = _bn_model()
    s1 = FlatParamStore(m1, torch.device("cpu"), torch.float32)
    sd = s1.to_state_dict()
    m2 = _bn_model()
    s2 = FlatParamStore(m2, torch.device("cpu"), torch.float32)
    s2.load_state_dict(sd)
    assert torch.allclose(s1.flat, s2.flat)
    flat = flatten_state_dict(sd, s1.spec)
    assert torch.allclose(flat, s1.flat)


def test_copy_from_flat_affects_forward():
    mlp = SimpleMLP(4, 8, 2)
    store = FlatParamStore(mlp, torch.device("cpu"), torch.float32)
    x = torch.randn(5, 4)
    out1 = store.model(x)
    store.copy_from_flat(torch.zeros_like(store.flat))
    out2 = store.model(x)
    assert torch.all(out2 == 0)
    assert not torch.allclose(out1, out2)


def test_resnet18_param_count_is_resnet18_sized():
    n = count_params(ResNet18(num_classes=10))
    assert 10.5e6 < n < 11.7e6  # ~11.2M like torchvision resnet18


def test_calculate_model_dimension():
    m = SimpleMLP(4, 8, 2)
    assert calculate_model_dimension(m) == sum(p.numel() for p in m.parameters())


def test_bf16_store():
    store = FlatParamStore(SimpleMLP(4, 8, 2), torch.device("cpu"), torch.bfloat16)
    assert store.flat.dtype == torch.bfloat16
    out = store.model(torch.randn(3, 4, dtype=torch.bfloat16))
    assert out.dtype == torch.bfloat16
