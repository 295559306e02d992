"""Evidential Trust: uncertainty-aware trust-weighted aggregation
(reference: murmura/aggregation/evidential_trust.py:25-469).

For each neighbor state: forward <= max_eval_samples local samples through the
(evidential) model -> Dirichlet alpha -> vacuity = K/S and accuracy; raw trust
= (1 - vacuity) * (w_a * acc + (1 - w_a)), with an exponential penalty
exp(-penalty_factor * (vacuity - tau_u)) when vacuity > tau_u
(evidential_trust.py:289-305); per-neighbor EMA smoothing
trust <- gamma_ema * new + (1 - gamma_ema) * old (:318-342); a TIGHTENING
acceptance threshold tau(t) = tau_base * (1 - gamma * exp(-kappa * t/T))
clamped to [0.05, tau_base] (:344-381); trust-weighted average of accepted
neighbors then self-blend alpha_self * own + (1 - alpha_self) * agg
(:194-212). Falls back to plain averaging when no eval context is available
(:150-153).

MI355X path: candidate scoring swaps flat states into a scratch model (no
deepcopy); vacuity/accuracy come from the fused evidential-stats kernel (K8);
EMA state is per-neighbor 0-dim device tensors; blending is one weighted-sum
launch.
"""

from __future__ import annotations

import math
from typing import Any, Dict, List, Optional

import torch
from torch import Tensor

from murmura_amd.aggregation.base import Aggregator, EvalContext, _to_float_list
from murmura_amd import ops


class EvidentialTrustAggregator(Aggregator):
    requires_eval_context = True

    def __init__(
        self,
        w_a: float = 0.5,
        tau_u: float = 0.5,
        penalty_factor: float = 1.0,
        gamma_ema: float = 0.7,
        tau_base: float = 0.3,
        gamma: float = 0.5,
        kappa: float = 1.0,
        alpha_self: float = 0.5,
        max_eval_samples: int = 100,
        total_rounds: int = 50,
    ):
        # defaults match the reference ctor (evidential_trust.py:43-58:
        # accuracy_weight=0.5, vacuity_threshold=0.5, trust_momentum=0.7,
        # trust_threshold=0.3, gamma=0.5, kappa=1.0, self_weight=0.5;
        # penalty exp(-(v - tau)) i.e. factor 1.0, :295-300)
        self.w_a = float(w_a)
        self.tau_u = float(tau_u)
        self.penalty_factor = float(penalty_factor)
        self.gamma_ema = float(gamma_ema)
        self.tau_base = float(tau_base)
        self.gamma = float(gamma)
        self.kappa = float(kappa)
        self.alpha_self = float(alpha_self)
        self.max_eval_samples = int(max_eval_samples)
        self.total_rounds = int(total_rounds)
        self._trust: Dict[int, Tensor] = {}  # neighbor id -> EMA trust (device scalar)
        self._trust_history: List[Tensor] = []
        self._acceptance_history: List[Tensor] = []

    def _threshold(self, round_num: int) -> float:
        t_frac = round_num / max(1, self.total_rounds)
        tau = self.tau_base * (1.0 - self.gamma * math.exp(-self.kappa * t_frac))
        return min(max(tau, 0.05), self.tau_base)

    def _raw_trust(self, vacuity: Tensor, acc: Tensor) -> Tensor:
        base = (1.0 - vacuity) * (self.w_a * acc + (1.0 - self.w_a))
        penalty = torch.exp(
            -self.penalty_factor * (vacuity - self.tau_u).clamp_min(0.0)
        )
        # trust clamped to [0, 1] (reference: evidential_trust.py:305)
        return (base * penalty).clamp(0.0, 1.0)

    def aggregate(
        self,
        node_id: int,
        own_state: Tensor,
        neighbor_states: Tensor,
        round_num: int = 0,
        eval_context: Optional[EvalContext] = None,
        neighbor_ids: Optional[List[int]] = None,
        **ctx: Any,
    ) -> Tensor:
        k = neighbor_states.shape[0]
        if k == 0:
            return own_state.clone()
        if eval_context is None:
            # plain averaging fallback (reference: evidential_trust.py:150-153)
            stacked = torch.cat([own_state.unsqueeze(0), neighbor_states], dim=0)
            w = torch.full(
                (k + 1,), 1.0 / (k + 1), device=own_state.device, dtype=torch.float32
            )
            return ops.weighted_sum(stacked, w)

        if neighbor_ids is None:
            neighbor_ids = list(range(k))

        # all k neighbors scored in ONE vmapped forward per data batch
        # (reference: deepcopy + forward per neighbor, evidential_trust.py:237)
        vac, acc = eval_context.evidential_scores(
            neighbor_states, self.max_eval_samples
        )
        new_trust = self._raw_trust(vac, acc)  # [k]
        prev = torch.stack(
            [
                self._trust.get(nid, torch.zeros((), device=new_trust.device)).to(
                    new_trust.device
                )
                for nid in neighbor_ids
            ]
        )
        has_prev = torch.tensor(
            [nid in self._trust for nid in neighbor_ids], device=new_trust.device
        )
        ema = self.gamma_ema * new_trust + (1.0 - self.gamma_ema) * prev
        trust_vec = torch.where(has_prev, ema, new_trust).float().view(-1)
        for i, nid in enumerate(neighbor_ids):
            self._trust[nid] = trust_vec[i].detach()

        tau = self._threshold(round_num)
        accept = trust_vec >= tau
        cnt = accept.sum()
        # trust-weighted average over accepted; if none accepted keep own state
        w_masked = trust_vec * accept.float()
        total = w_masked.sum()
        none_accepted = (cnt == 0).float()
        w_nbr = w_masked / total.clamp(min=1e-12)
        # self-blend; when none accepted the blend collapses to own state
        alpha_eff = self.alpha_self + (1.0 - self.alpha_self) * none_accepted
        stacked = torch.cat([own_state.unsqueeze(0), neighbor_states], dim=0)
        w_full = torch.cat(
            [alpha_eff.view(1), (1.0 - alpha_eff) * w_nbr]
        )
        out = ops.weighted_sum(stacked, w_full)

        self._trust_history.append(trust_vec.mean())
        self._acceptance_history.append(accept.float().mean())
        return out

    def get_statistics(self) -> Dict[str, Any]:
        return {
            "mean_trust": _to_float_list(self._trust_history),
            "acceptance_rates": _to_float_list(self._acceptance_history),
            "per_neighbor_trust": {
                nid: float(t.detach().cpu()) for nid, t in self._trust.items()
            },
        }
