import math

import pytest
import torch
from torch.utils.data import DataLoader, TensorDataset

from murmura_amd.aggregation import (
    BALANCEAggregator,
    EvidentialTrustAggregator,
    FedAvgAggregator,
    KrumAggregator,
    SketchguardAggregator,
    UBARAggregator,
)
from murmura_amd.aggregation.base import EvalContext, accept_weights
from murmura_amd.core.flat import FlatParamStore
from murmura_amd.models import SimpleMLP


P = 64


def _states(m, scale=1.0, seed=0):
    g = torch.Generator().manual_seed(seed)
    return torch.randn(m, P, generator=g) * scale


# --------------------------------------------------------------- fedavg
def test_fedavg_is_mean():
    own = torch.ones(P)
    nbrs = torch.stack([torch.full((P,), 3.0), torch.full((P,), 5.0)])
    out = FedAvgAggregator().aggregate(0, own, nbrs)
    assert torch.allclose(out, torch.full((P,), 3.0))


def test_fedavg_no_neighbors():
    own = torch.randn(P)
    out = FedAvgAggregator().aggregate(0, own, own.new_zeros((0, P)))
    assert torch.equal(out, own)


# --------------------------------------------------------------- krum
def test_krum_rejects_outlier():
    own = torch.randn(P) * 0.01
    inliers = _states(4, scale=0.01, seed=1)
    outlier = torch.full((P,), 50.0)
    nbrs = torch.cat([inliers, outlier.unsqueeze(0)])
    out = KrumAggregator(num_compromised=1).aggregate(0, own, nbrs)
    # selected state must be one of the inlier states (or own), not the outlier
    assert out.abs().max() < 1.0


def test_krum_fallback_when_constraint_violated():
    own = torch.randn(P)
    nbrs = _states(3, seed=2)
    # m=4, c=1 -> c >= (m-2)/2 = 1 -> fallback to own
    agg = KrumAggregator(num_compromised=1)
    out = agg.aggregate(0, own, nbrs)
    assert torch.equal(out, own)
    assert agg.get_statistics()["fallbacks"] == 1


def test_krum_returns_state_verbatim():
    own = torch.zeros(P)
    nbrs = _states(5, scale=0.1, seed=3)
    out = KrumAggregator(num_compromised=0).aggregate(0, own, nbrs)
    stacked = torch.cat([own.unsqueeze(0), nbrs])
    assert any(torch.equal(out, stacked[i]) for i in range(6))


# --------------------------------------------------------------- balance
def test_balance_accepts_close_rejects_far():
    own = torch.ones(P)
    close = own + 0.01
    far = own * 100.0
    nbrs = torch.stack([close, far])
    agg = BALANCEAggregator(gamma=0.5, kappa=1.0, alpha=0.5, total_rounds=50)
    out = agg.aggregate(0, own, nbrs, round_num=0)
    # accepted = {close} -> out = 0.5*own + 0.5*close
    assert torch.allclose(out, 0.5 * own + 0.5 * close, atol=1e-5)


def test_balance_threshold_decays():
    agg = BALANCEAggregator(gamma=2.0, kappa=1.0, total_rounds=10)
    assert agg._decay(0) == pytest.approx(2.0)
    assert agg._decay(10) == pytest.approx(2.0 * math.exp(-1.0))


def test_balance_fallback_accepts_closest():
    own = torch.ones(P)
    n1 = own * 50.0
    n2 = own * 100.0
    agg = BALANCEAggregator(gamma=0.01, alpha=0.5, min_neighbors=1)
    out = agg.aggregate(0, own, torch.stack([n1, n2]), round_num=0)
    assert torch.allclose(out, 0.5 * own + 0.5 * n1, atol=1e-4)


def test_accept_weights_branchless():
    dists = torch.tensor([1.0, 2.0, 3.0])
    mask = torch.tensor([True, True, False])
    w = accept_weights(mask, dists, 1)
    assert torch.allclose(w, torch.tensor([0.5, 0.5, 0.0]))
    none = torch.tensor([False, False, False])
    w2 = accept_weights(none, dists, 1)
    assert torch.allclose(w2, torch.tensor([1.0, 0.0, 0.0]))


# --------------------------------------------------------------- sketchguard
def test_sketchguard_filters_like_balance_but_in_sketch_space():
    own = torch.ones(P)
    close = own + 0.01
    far = own * 100.0
    agg = SketchguardAggregator(model_dim=P, sketch_size=32, gamma=0.5, alpha=0.5)
    out = agg.aggregate(0, own, torch.stack([close, far]), round_num=0)
    assert torch.allclose(out, 0.5 * own + 0.5 * close, atol=1e-4)


def test_sketchguard_sketch_shared_tables():
    a = SketchguardAggregator(model_dim=P, sketch_size=16, network_seed=7)
    b = SketchguardAggregator(model_dim=P, sketch_size=16, network_seed=7)
    x = torch.randn(P)
    assert torch.allclose(a.get_sketch(x), b.get_sketch(x))


def test_sketchguard_accepts_precomputed_sketches():
    own = torch.ones(P)
    nbrs = torch.stack([own + 0.01, own * 100.0])
    agg = SketchguardAggregator(model_dim=P, sketch_size=32, gamma=0.5)
    sk = torch.stack([agg.get_sketch(nbrs[0]), agg.get_sketch(nbrs[1])])
    out = agg.aggregate(0, own, nbrs, round_num=0, neighbor_sketches=sk)
    ref = SketchguardAggregator(model_dim=P, sketch_size=32, gamma=0.5).aggregate(
        0, own, nbrs, round_num=0
    )
    assert torch.allclose(out, ref, atol=1e-5)


# --------------------------------------------------------------- ubar / evidential
def _eval_ctx(evidential=False, in_features=8, num_classes=3):
    model = SimpleMLP(in_features, 16, num_classes)
    store = FlatParamStore(model, torch.device("cpu"), torch.float32)
    g = torch.Generator().manual_seed(0)
    x = torch.randn(64, in_features, generator=g)
    y = torch.randint(0, num_classes, (64,), generator=g)
    loader = DataLoader(TensorDataset(x, y), batch_size=32)
    return EvalContext(store, loader, torch.device("cpu"), evidential), store


def test_ubar_rejects_garbage_state():
    torch.manual_seed(0)
    ctx, store = _eval_ctx()
    p = store.spec.total_numel
    # own = trained-ish random state; garbage = huge weights (terrible loss)
    g = torch.Generator().manual_seed(1)
    own = torch.randn(p, generator=g) * 0.1
    good = own + torch.randn(p, generator=g) * 0.01
    # large random weights => confidently-wrong predictions => loss >> own's
    garbage = torch.randn(p, generator=g) * 50.0
    batch = ctx.next_batch()
    assert ctx.loss_on_batch(garbage, batch) > ctx.loss_on_batch(own, batch)
    nbrs = torch.stack([good, garbage])
    agg = UBARAggregator(rho=1.0, alpha=0.5)
    out = agg.aggregate(0, own, nbrs, round_num=0, eval_context=ctx)
    # garbage excluded by the performance filter; blend of own and good only
    expect = 0.5 * own + 0.5 * good
    assert torch.allclose(out, expect, atol=1e-5)


def test_ubar_requires_eval_context():
    with pytest.raises(ValueError):
        UBARAggregator().aggregate(0, torch.randn(P), _states(2), 0)


def test_ubar_stage1_keeps_rho_fraction():
    torch.manual_seed(0)
    ctx, store = _eval_ctx()
    p = store.spec.total_numel
    own = torch.randn(p) * 0.1
    nbrs = own.unsqueeze(0) + torch.randn(5, p) * 0.01
    agg = UBARAggregator(rho=0.4, alpha=0.5)
    agg.aggregate(0, own, nbrs, round_num=0, eval_context=ctx)
    assert agg.get_statistics()["stage1_kept"] == [2.0]  # ceil(0.4*5)


def test_evidential_trust_plain_average_without_ctx():
    own = torch.ones(P)
    nbrs = torch.stack([torch.full((P,), 4.0)])
    out = EvidentialTrustAggregator().aggregate(0, own, nbrs, round_num=0)
    assert torch.allclose(out, torch.full((P,), 2.5))


def test_evidential_trust_raw_trust_semantics():
    """Trust = (1 - vacuity) * (w_a * acc + (1 - w_a)) with an exponential
    penalty above tau_u: decreasing in vacuity, increasing in accuracy
    (reference: evidential_trust.py:289-305). NOTE the formula is
    vacuity-driven — a confidently-wrong model scores high, which is exactly
    why the reference's evidential_trust collapses under gaussian attack
    (28.13% final acc)."""
    agg = EvidentialTrustAggregator(w_a=0.7, tau_u=0.5, penalty_factor=5.0)
    t = lambda v, a: agg._raw_trust(torch.tensor(v), torch.tensor(a)).item()
    assert t(0.1, 0.9) > t(0.6, 0.9)  # higher vacuity -> lower trust
    assert t(0.1, 0.9) > t(0.1, 0.2)  # higher accuracy -> higher trust
    # penalty kicks in only above tau_u
    assert t(0.49, 0.5) > t(0.51, 0.5)
    assert abs(t(0.3, 0.5) - (0.7 * (0.7 * 0.5 + 0.3))) < 1e-5


def test_evidential_trust_uses_eval_and_blends():
    torch.manual_seed(0)
    ctx, store = _eval_ctx(evidential=True)
    p = store.spec.total_numel
    own = torch.randn(p) * 0.1
    good = own + torch.randn(p) * 0.01
    agg = EvidentialTrustAggregator(total_rounds=50)
    out = agg.aggregate(
        0, own, torch.stack([good]), round_num=25,
        eval_context=ctx, neighbor_ids=[1],
    )
    stats = agg.get_statistics()
    assert 0.0 <= stats["per_neighbor_trust"][1] <= 1.0
    assert out.shape == own.shape
    # output stays within the convex hull of own/neighbor coordinates
    lo = torch.minimum(own, good) - 1e-4
    hi = torch.maximum(own, good) + 1e-4
    assert torch.all(out >= lo) and torch.all(out <= hi)


def test_evidential_trust_ema():
    torch.manual_seed(0)
    ctx, store = _eval_ctx(evidential=True)
    p = store.spec.total_numel
    own = torch.randn(p) * 0.1
    nbr = (own + 0.01).unsqueeze(0)
    agg = EvidentialTrustAggregator(gamma_ema=0.5)
    agg.aggregate(0, own, nbr, 0, eval_context=ctx, neighbor_ids=[1])
    t1 = agg.get_statistics()["per_neighbor_trust"][1]
    agg.aggregate(0, own, nbr, 1, eval_context=ctx, neighbor_ids=[1])
    t2 = agg.get_statistics()["per_neighbor_trust"][1]
    # EMA: same raw trust each round keeps value roughly stable
    assert abs(t2 - t1) < 0.5


# --------------------------------------------------------------- dict compat
def test_compat_average_states_copies_nonfloat():
    import torch.nn as nn

    from murmura_amd.aggregation.compat import (
        average_states,
        compute_model_distance,
        flatten_model_state,
    )

    m = nn.Sequential(nn.Linear(4, 4), nn.BatchNorm1d(4))
    m(torch.randn(8, 4))  # tick num_batches_tracked
    s1 = {k: v.clone() for k, v in m.state_dict().items()}
    s2 = {k: v.clone() * 3 if torch.is_floating_point(v) else v.clone()
          for k, v in s1.items()}
    avg = average_states([s1, s2])
    assert torch.allclose(avg["0.weight"], 2 * s1["0.weight"])
    # non-float buffer copied, not averaged
    assert avg["1.num_batches_tracked"].dtype == torch.int64
    assert avg["1.num_batches_tracked"].item() == s1["1.num_batches_tracked"].item()
    # distance and flatten agree
    d = compute_model_distance(s1, s2)
    f1, f2 = flatten_model_state(s1), flatten_model_state(s2)
    assert d == pytest.approx((f1 - f2).norm().item(), rel=1e-5)


def test_batched_scoring_matches_serial():
    """EvalContext.losses_on_batch / evidential_scores (one vmapped forward
    for all candidates) must equal the per-candidate serial path."""
    from murmura_amd.aggregation.base import EvalContext
    from murmura_amd.core.flat import FlatParamStore
    from murmura_amd.models.evidential import EvidentialHARClassifier
    from torch.utils.data import DataLoader, TensorDataset

    torch.manual_seed(0)
    model = EvidentialHARClassifier(input_dim=20, hidden_dims=(16,), num_classes=4)
    store = FlatParamStore(model, torch.device("cpu"), torch.float32)
    ds = TensorDataset(torch.randn(64, 20), torch.randint(0, 4, (64,)))
    ctx = EvalContext(store, DataLoader(ds, batch_size=32), torch.device("cpu"),
                      evidential=True)
    base = store.flat.clone()
    states = torch.stack([base + 0.05 * torch.randn_like(base) for _ in range(3)])
    batch = ctx.next_batch()
    batched = ctx.losses_on_batch(states, batch)
    serial = torch.stack([ctx.loss_on_batch(states[i], batch) for i in range(3)])
    assert ctx._vmap_ok is True  # the fast path actually ran
    assert torch.allclose(batched, serial, atol=1e-5)
    v, a = ctx.evidential_scores(states, 64)
    pairs = [ctx.evidential_score(states[i], 64) for i in range(3)]
    assert torch.allclose(v, torch.stack([p[0] for p in pairs]), atol=1e-5)
    assert torch.allclose(a, torch.stack([p[1] for p in pairs]), atol=1e-5)
