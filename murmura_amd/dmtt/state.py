"""DMTT per-node trust state (reference: murmura/dmtt/state.py:22-142).

Scalar O(N)-per-round host-side math (stays on host by
design). All state is per-peer:

- link reliability EMA:   c_hat <- (1 - rho) * c_hat + rho * ack   (init 0.5)
- Beta-evidence source trust:  alpha <- lam * alpha + w_d * d + w_c * c
                               beta  <- lam * beta  + w_x * x      (floor 0.01,
                               prior Beta(1, 1))
- topology trust: T = R * exp(-eta * max(0, U - tau_U)) with R = a/(a+b) and
  U = posterior std of the Beta
- model score: s = (1 - vac) * (w_a * acc + (1 - w_a)), exp penalty above tau_u
- collaborator score: q = l1*s + l2*T + l3*c_hat - l4*c_comm
- top_b: rank candidates by q, return <= B
"""

from __future__ import annotations

import math
from typing import Dict, List, Sequence

from murmura_amd.config.schema import DMTTConfig


class DMTTNodeState:
    """Not thread-safe; use from a single rank's process only (the reference
    documents the same constraint, dmtt/state.py:26)."""

    def __init__(self, node_id: int, num_nodes: int, cfg: DMTTConfig):
        self.node_id = node_id
        self.num_nodes = num_nodes
        self.cfg = cfg
        self.link_reliability: Dict[int, float] = {
            j: 0.5 for j in range(num_nodes) if j != node_id
        }
        self.alpha: Dict[int, float] = {j: 1.0 for j in range(num_nodes) if j != node_id}
        self.beta: Dict[int, float] = {j: 1.0 for j in range(num_nodes) if j != node_id}
        self.model_scores: Dict[int, float] = {}

    # ---------------------------------------------------------------- links
    def update_link_reliability(self, peer: int, ack: bool) -> None:
        c = self.link_reliability[peer]
        self.link_reliability[peer] = (1.0 - self.cfg.rho) * c + self.cfg.rho * (
            1.0 if ack else 0.0
        )

    # ---------------------------------------------------------------- Beta evidence
    def update_topology_evidence(self, peer: int, d: float, c: float, x: float) -> None:
        """d = direct matches, c = corroborations, x = contradictions."""
        lam = self.cfg.lambda_forget
        a = lam * self.alpha[peer] + self.cfg.w_d * d + self.cfg.w_c * c
        b = lam * self.beta[peer] + self.cfg.w_x * x
        self.alpha[peer] = max(a, 0.01)
        self.beta[peer] = max(b, 0.01)

    def topology_trust(self, peer: int) -> float:
        a, b = self.alpha[peer], self.beta[peer]
        r = a / (a + b)
        u = math.sqrt(a * b / ((a + b) ** 2 * (a + b + 1.0)))
        return r * math.exp(-self.cfg.eta * max(0.0, u - self.cfg.tau_U))

    # ---------------------------------------------------------------- model score
    def model_score(self, vacuity: float, accuracy: float) -> float:
        # a poisoned state can drive the evidential head to inf/NaN alpha
        # (e.g. sigma-30 Gaussian weights); a NaN score would poison the TopB
        # sort (NaN comparisons are all False), silently keeping attackers in
        # C_i^t — treat non-finite evidence as zero trust
        if not (math.isfinite(vacuity) and math.isfinite(accuracy)):
            return 0.0
        s = (1.0 - vacuity) * (self.cfg.w_a * accuracy + (1.0 - self.cfg.w_a))
        if vacuity > self.cfg.tau_u:
            s *= math.exp(-self.cfg.eta * (vacuity - self.cfg.tau_u))
        return max(0.0, min(1.0, s))

    def record_model_score(self, peer: int, vacuity: float, accuracy: float) -> None:
        self.model_scores[peer] = self.model_score(vacuity, accuracy)

    # ---------------------------------------------------------------- selection
    def collaborator_score(self, peer: int, comm_cost: float = 0.0) -> float:
        c = self.cfg
        s = self.model_scores.get(peer, 0.5)
        return (
            c.lambda1 * s
            + c.lambda2 * self.topology_trust(peer)
            + c.lambda3 * self.link_reliability[peer]
            - c.lambda4 * comm_cost
        )

    def top_b(
        self, candidates: Sequence[int], budget: int | None = None,
        comm_costs: Dict[int, float] | None = None,
    ) -> List[int]:
        """Rank candidate peers by collaborator score, return the top <= B."""
        b = budget if budget is not None else self.cfg.budget_B
        costs = comm_costs or {}
        ranked = sorted(
            (j for j in candidates if j != self.node_id),
            key=lambda j: self.collaborator_score(j, costs.get(j, 0.0)),
            reverse=True,
        )
        return ranked[:b]
