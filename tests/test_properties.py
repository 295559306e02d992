"""Property-based invariants (hypothesis) for partitioners, topologies and
aggregation weight helpers."""

import numpy as np
import torch
from hypothesis import given, settings
from hypothesis import strategies as st

from murmura_amd.aggregation.base import accept_weights
from murmura_amd.data.partitioners import dirichlet_partition, iid_partition
from murmura_amd.topology.generators import create_topology


@settings(max_examples=30, deadline=None)
@given(n=st.integers(2, 200), clients=st.integers(1, 12), seed=st.integers(0, 1000))
def test_iid_partition_is_a_partition(n, clients, seed):
    parts = iid_partition(n, clients, seed=seed)
    flat = sorted(i for p in parts for i in p)
    assert flat == list(range(n))
    assert len(parts) == clients


@settings(max_examples=20, deadline=None)
@given(
    n=st.integers(20, 300),
    classes=st.integers(2, 8),
    clients=st.integers(2, 8),
    alpha=st.floats(0.05, 10.0),
    seed=st.integers(0, 100),
)
def test_dirichlet_partition_is_a_partition(n, classes, clients, alpha, seed):
    rng = np.random.default_rng(seed)
    labels = rng.integers(0, classes, n).tolist()
    parts = dirichlet_partition(labels, clients, alpha=alpha, seed=seed)
    flat = sorted(i for p in parts for i in p)
    assert flat == list(range(n))


@settings(max_examples=30, deadline=None)
@given(
    n=st.integers(2, 40),
    kind=st.sampled_from(["ring", "fully", "erdos", "k-regular"]),
    p=st.floats(0.0, 1.0),
    k=st.integers(2, 10),
    seed=st.integers(0, 1000),
)
def test_topology_invariants(n, kind, p, k, seed):
    import warnings

    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        t = create_topology(kind, n, p=p, k=k, seed=seed)
    # adjacency consistent with edge list; no self-loops; symmetric
    for i, j in t.edges:
        assert i < j
        assert j in t.neighbors[i] and i in t.neighbors[j]
    deg_sum = sum(t.degree(i) for i in range(n))
    assert deg_sum == 2 * len(t.edges)
    if kind in ("ring", "fully", "k-regular") and n > 2:
        assert t.is_connected()
    if kind == "erdos" and n > 1:
        assert all(t.degree(i) >= 1 for i in range(n))


@settings(max_examples=40, deadline=None)
@given(
    k=st.integers(1, 12),
    seed=st.integers(0, 1000),
    thresh=st.floats(0.0, 3.0),
)
def test_accept_weights_is_distribution(k, seed, thresh):
    g = torch.Generator().manual_seed(seed)
    dists = torch.rand(k, generator=g) * 2.0
    mask = dists <= thresh
    w = accept_weights(mask, dists, min_neighbors=1)
    assert torch.all(w >= 0)
    assert abs(w.sum().item() - 1.0) < 1e-5
    if mask.any():
        # only accepted neighbors carry weight
        assert torch.all(w[~mask] == 0)
    else:
        # fallback: all weight on the closest
        assert w[dists.argmin()].item() == 1.0


@settings(max_examples=20, deadline=None)
@given(n=st.integers(2, 12), seed=st.integers(0, 500), r=st.integers(0, 10))
def test_mobility_positions_and_determinism(n, seed, r):
    from murmura_amd.topology.dynamic import MobilityModel

    a = MobilityModel(n, area_size=50.0, comm_range=20.0, seed=seed)
    b = MobilityModel(n, area_size=50.0, comm_range=20.0, seed=seed)
    pa, pb = a.positions_at(r), b.positions_at(r)
    assert (pa == pb).all()
    assert (pa >= 0).all() and (pa < 50.0).all()
    ta, tb = a.topology_at(r), b.topology_at(r)
    assert ta.edges == tb.edges
    assert all(ta.degree(i) >= 1 for i in range(n))  # ensure_connected default


# ------------------------------------------------------- aggregation invariants
from hypothesis import given, settings, strategies as st


@settings(max_examples=25, deadline=None)
@given(st.integers(2, 9), st.integers(3, 40), st.randoms(use_true_random=False))
def test_fedavg_permutation_invariant(k, p, rnd):
    """FedAvg must not depend on neighbor order (gossip arrival order is
    nondeterministic in the reference's ZMQ transport)."""
    import torch

    from murmura_amd.aggregation import FedAvgAggregator

    g = torch.Generator().manual_seed(rnd.randint(0, 2**31))
    own = torch.randn(p, generator=g)
    nbrs = torch.randn(k, p, generator=g)
    perm = torch.randperm(k, generator=g)
    a = FedAvgAggregator().aggregate(0, own, nbrs)
    b = FedAvgAggregator().aggregate(0, own, nbrs[perm])
    assert torch.allclose(a, b, atol=1e-5)


@settings(max_examples=25, deadline=None)
@given(st.integers(4, 9), st.integers(3, 40), st.randoms(use_true_random=False))
def test_krum_selects_a_member(k, p, rnd):
    """Krum returns one of {own, neighbors} verbatim — never a blend."""
    import torch

    from murmura_amd.aggregation import KrumAggregator

    g = torch.Generator().manual_seed(rnd.randint(0, 2**31))
    own = torch.randn(p, generator=g)
    nbrs = torch.randn(k, p, generator=g)
    out = KrumAggregator(num_compromised=1).aggregate(0, own, nbrs)
    members = [own] + [nbrs[i] for i in range(k)]
    assert any(torch.equal(out, m) for m in members)


@settings(max_examples=25, deadline=None)
@given(st.integers(2, 9), st.integers(3, 40), st.randoms(use_true_random=False))
def test_balance_output_in_convex_hull_bounds(k, p, rnd):
    """BALANCE's alpha-blend of own + accepted mean stays inside the
    coordinate-wise min/max envelope of its inputs."""
    import torch

    from murmura_amd.aggregation import BALANCEAggregator

    g = torch.Generator().manual_seed(rnd.randint(0, 2**31))
    own = torch.randn(p, generator=g)
    nbrs = own.unsqueeze(0) + 0.1 * torch.randn(k, p, generator=g)
    out = BALANCEAggregator().aggregate(0, own, nbrs, round_num=0)
    allv = torch.cat([own.unsqueeze(0), nbrs])
    assert bool((out <= allv.max(dim=0).values + 1e-5).all())
    assert bool((out >= allv.min(dim=0).values - 1e-5).all())
