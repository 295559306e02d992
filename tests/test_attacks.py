import torch

from murmura_amd.attacks import (
    DirectedDeviationAttack,
    GaussianAttack,
    TopologyLiarAttack,
    select_compromised,
)


def test_select_compromised_count_and_determinism():
    assert select_compromised(10, 0.0, 1) == []
    assert len(select_compromised(10, 0.2, 1)) == 2
    assert len(select_compromised(10, 0.25, 1)) == 2  # floor (reference: gaussian.py:37)
    assert len(select_compromised(10, 0.05, 1)) == 1  # at least 1 when pct > 0
    assert len(select_compromised(10, 0.01, 1)) == 1  # at least 1
    assert select_compromised(10, 0.3, 7) == select_compromised(10, 0.3, 7)
    assert select_compromised(10, 1.0, 1) == list(range(10))


def test_gaussian_attack():
    atk = GaussianAttack(10, 0.2, noise_std=2.0, seed=42)
    assert len(atk.get_compromised_nodes()) == 2
    x = torch.zeros(10_000)
    nid = atk.get_compromised_nodes()[0]
    out = atk.apply_attack(nid, x, round_num=0)
    assert abs(out.std().item() - 2.0) < 0.1
    # deterministic per (node, round); different across rounds
    out2 = atk.apply_attack(nid, x, round_num=0)
    out3 = atk.apply_attack(nid, x, round_num=1)
    assert torch.equal(out, out2)
    assert not torch.equal(out, out3)
    # input not mutated
    assert torch.all(x == 0)


def test_directed_deviation():
    atk = DirectedDeviationAttack(10, 0.2, deviation_factor=-5.0, seed=42)
    x = torch.randn(100)
    out = atk.apply_attack(atk.get_compromised_nodes()[0], x, 0)
    assert torch.allclose(out, -5.0 * x)


def test_topology_liar_claims():
    atk = TopologyLiarAttack(10, 0.3, seed=1)
    comp = atk.get_compromised_nodes()
    assert len(comp) == 3
    liar = comp[0]
    true_nbrs = [5, 6]
    claims = atk.get_false_claims(liar, true_nbrs, 0)
    for c in comp:
        if c != liar:
            assert c in claims
    for n in true_nbrs:
        assert n in claims
    assert liar not in claims


def test_topology_liar_wraps_model_attack():
    inner = DirectedDeviationAttack(10, 0.3, deviation_factor=-2.0, seed=1)
    atk = TopologyLiarAttack(10, 0.3, seed=1, model_attack=inner)
    x = torch.randn(50)
    out = atk.apply_attack(atk.get_compromised_nodes()[0], x, 0)
    assert torch.allclose(out, -2.0 * x)
    # without inner attack the state passes through unchanged
    plain = TopologyLiarAttack(10, 0.3, seed=1)
    assert torch.allclose(plain.apply_attack(0, x, 0), x)
