"""Torch reference ops vs naive loop implementations (the oracle's oracle)."""

import math

import pytest
import torch

from murmura_amd.ops import reference as ref


def test_weighted_sum_matches_loop():
    m, p = 5, 137
    x = torch.randn(m, p)
    w = torch.rand(m)
    out = ref.weighted_sum(x, w)
    naive = sum(w[i] * x[i] for i in range(m))
    assert torch.allclose(out, naive, atol=1e-5)


def test_weighted_sum_out_arg():
    x = torch.randn(3, 10)
    w = torch.ones(3) / 3
    out = torch.empty(10)
    r = ref.weighted_sum(x, w, out)
    assert r is out
    assert torch.allclose(out, x.mean(0), atol=1e-6)


def test_pairwise_l2_matches_loop():
    m, p = 6, 91
    x = torch.randn(m, p)
    d = ref.pairwise_l2(x)
    for i in range(m):
        for j in range(m):
            expect = (x[i] - x[j]).norm()
            assert abs(d[i, j].item() - expect.item()) < 1e-3
    assert torch.allclose(d, d.t(), atol=1e-4)
    assert torch.all(d.diagonal().abs() < 1e-3)


def test_row_norms_and_dists_to():
    x = torch.randn(4, 50)
    own = torch.randn(50)
    assert torch.allclose(ref.row_norms(x), x.norm(dim=1), atol=1e-5)
    d = ref.l2_dists_to(own, x)
    for i in range(4):
        assert abs(d[i].item() - (x[i] - own).norm().item()) < 1e-4


def test_krum_scores_and_select():
    # 4 inliers near 0, one far outlier; krum must not select the outlier
    m, p = 5, 20
    x = torch.randn(m, p) * 0.01
    x[3] += 100.0
    d2 = ref.pairwise_sq_dists(x)
    scores = ref.krum_scores(d2, num_compromised=1)
    assert scores.argmax().item() == 3
    sel = ref.krum_select(d2, 1).item()
    assert sel != 3
    # score definition: sum of (m - c - 2) smallest distances to others
    d = d2.sqrt()
    i = 0
    others = sorted(d[i, j].item() for j in range(m) if j != i)
    expect = sum(others[: m - 1 - 2])
    assert abs(scores[0].item() - expect) < 1e-3


def test_count_sketch_matches_bincount():
    import numpy as np

    p, s = 500, 37
    x = torch.randn(p)
    h, sg = ref.make_sketch_tables(p, s, seed=3, device=torch.device("cpu"))
    out = ref.count_sketch(x, h, sg, s)
    expect = np.bincount(
        h.numpy(), weights=(sg.numpy() * x.numpy()), minlength=s
    )
    assert torch.allclose(out, torch.from_numpy(expect).float(), atol=1e-4)


def test_count_sketch_batched():
    p, s, m = 200, 29, 3
    x = torch.randn(m, p)
    h, sg = ref.make_sketch_tables(p, s, seed=5, device=torch.device("cpu"))
    out = ref.count_sketch(x, h, sg, s)
    for i in range(m):
        assert torch.allclose(out[i], ref.count_sketch(x[i], h, sg, s), atol=1e-5)


def test_sketch_preserves_l2_roughly():
    # Count-Sketch is an unbiased L2 estimator: with sketch_size >> 1 the
    # distance between sketches approximates the true distance
    p, s = 20000, 4096
    a, b = torch.randn(p), torch.randn(p)
    h, sg = ref.make_sketch_tables(p, s, seed=1, device=torch.device("cpu"))
    true = (a - b).norm().item()
    est = (ref.count_sketch(a, h, sg, s) - ref.count_sketch(b, h, sg, s)).norm().item()
    assert abs(est - true) / true < 0.25


def test_sgd_step():
    p = torch.ones(10)
    g = torch.full((10,), 2.0)
    ref.sgd_step(p, g, lr=0.5)
    assert torch.allclose(p, torch.zeros(10))


def test_ce_loss_acc():
    logits = torch.tensor([[10.0, 0.0], [0.0, 10.0], [10.0, 0.0]])
    targets = torch.tensor([0, 1, 1])
    loss, correct = ref.ce_loss_acc(logits, targets)
    assert correct.item() == 2
    expect = torch.nn.functional.cross_entropy(logits, targets, reduction="sum")
    assert torch.allclose(loss, expect)


def test_evidential_stats():
    b, k = 8, 5
    logits = torch.randn(b, k)
    targets = torch.randint(0, k, (b,))
    v, e, s, c = ref.evidential_stats(logits, targets)
    alpha = torch.nn.functional.softplus(logits) + 1
    S = alpha.sum(1)
    assert torch.allclose(v, (k / S).sum(), atol=1e-4)
    assert torch.allclose(s, S.sum(), atol=1e-3)
    p = alpha / S.unsqueeze(1)
    ent = -(p * p.log()).sum(1).sum()
    assert torch.allclose(e, ent, atol=1e-3)
    assert c.item() == (alpha.argmax(1) == targets).sum().item()
    # vacuity bounds: K/S with S >= K => vacuity in (0, 1]
    assert 0 < (v / b).item() <= 1.0


def test_gaussian_inject_stats_and_determinism():
    x = torch.zeros(100_000)
    a = ref.gaussian_inject(x, noise_std=3.0, seed=11, offset=2)
    b = ref.gaussian_inject(x, noise_std=3.0, seed=11, offset=2)
    c = ref.gaussian_inject(x, noise_std=3.0, seed=11, offset=3)
    assert torch.equal(a, b)
    assert not torch.equal(a, c)
    assert abs(a.std().item() - 3.0) < 0.05
    assert abs(a.mean().item()) < 0.05


def test_scale_inject():
    x = torch.randn(50)
    assert torch.allclose(ref.scale_inject(x, -5.0), -5.0 * x)


def test_sketch_tables_grouped_structure():
    """make_sketch_tables: bins constant per aligned group of 8, signs per
    element, deterministic per seed, and sketch distances stay unbiased
    enough to preserve orderings (the MI355X-native table layout —
    ops/reference.py)."""
    import torch

    from murmura_amd.ops import reference as ref

    P, S = 10_000, 1000
    h, s = ref.make_sketch_tables(P, S, seed=3, device=torch.device("cpu"))
    h2, s2 = ref.make_sketch_tables(P, S, seed=3, device=torch.device("cpu"))
    assert torch.equal(h, h2) and torch.equal(s, s2)  # deterministic
    ng = P // 8
    hg = h[: ng * 8].view(ng, 8)
    assert bool((hg == hg[:, :1]).all())  # group-constant bins
    # signs vary within groups (per-element)
    sg = s[: ng * 8].view(ng, 8)
    assert not bool((sg == sg[:, :1]).all())
    # distance-ordering sanity: a far vector must sketch farther than a near one
    base = torch.randn(P)
    near = base + 0.01 * torch.randn(P)
    far = base + 10.0 * torch.randn(P)
    sk = lambda v: ref.count_sketch(v.unsqueeze(0), h, s, S)[0]
    d_near = (sk(base) - sk(near)).norm()
    d_far = (sk(base) - sk(far)).norm()
    assert d_far > 10 * d_near
