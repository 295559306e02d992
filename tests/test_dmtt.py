import math

import pytest

from murmura_amd.config.schema import DMTTConfig
from murmura_amd.dmtt.state import DMTTNodeState


def _state(**kw):
    return DMTTNodeState(0, 5, DMTTConfig(**kw))


def test_link_reliability_ema():
    s = _state(rho=0.1)
    assert s.link_reliability[1] == 0.5  # init
    s.update_link_reliability(1, ack=True)
    assert s.link_reliability[1] == pytest.approx(0.55)
    s.update_link_reliability(1, ack=False)
    assert s.link_reliability[1] == pytest.approx(0.495)


def test_beta_evidence_and_topology_trust():
    s = _state(lambda_forget=0.9, w_d=1.0, w_x=1.0, eta=5.0, tau_U=0.3)
    # prior Beta(1,1): R = 0.5, high uncertainty
    t0 = s.topology_trust(1)
    # consistent honest evidence raises trust
    for _ in range(10):
        s.update_topology_evidence(1, d=3.0, c=0.0, x=0.0)
    t_honest = s.topology_trust(1)
    # consistent contradictions tank trust
    for _ in range(10):
        s.update_topology_evidence(2, d=0.0, c=0.0, x=3.0)
    t_liar = s.topology_trust(2)
    assert t_honest > t0 > t_liar
    assert t_liar < 0.2


def test_beta_floor():
    s = _state(lambda_forget=0.0)
    s.update_topology_evidence(1, d=0.0, c=0.0, x=0.0)
    assert s.alpha[1] == 0.01 and s.beta[1] == 0.01


def test_model_score_penalty():
    s = _state(w_a=0.7, tau_u=0.5, eta=5.0)
    good = s.model_score(vacuity=0.1, accuracy=0.9)
    vac = s.model_score(vacuity=0.8, accuracy=0.9)
    assert good > vac
    assert good == pytest.approx(0.9 * (0.7 * 0.9 + 0.3))


def test_collaborator_score_and_top_b():
    s = _state(budget_B=2, lambda1=0.4, lambda2=0.3, lambda3=0.2, lambda4=0.1)
    s.record_model_score(1, vacuity=0.1, accuracy=0.9)
    s.record_model_score(2, vacuity=0.1, accuracy=0.5)
    s.record_model_score(3, vacuity=0.9, accuracy=0.1)
    for _ in range(5):
        s.update_topology_evidence(1, d=2, c=0, x=0)
        s.update_topology_evidence(3, d=0, c=0, x=2)
    top = s.top_b([1, 2, 3])
    assert len(top) == 2
    assert top[0] == 1
    assert 3 not in top
    # comm cost lowers the score
    lo = s.collaborator_score(1, comm_cost=10.0)
    hi = s.collaborator_score(1, comm_cost=0.0)
    assert hi > lo


def test_top_b_excludes_self_and_respects_budget():
    s = _state(budget_B=10)
    top = s.top_b([0, 1, 2], budget=1)
    assert 0 not in top and len(top) == 1
