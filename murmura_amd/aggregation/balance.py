"""BALANCE: distance-filtered averaging with an adaptive decaying threshold
(reference: murmura/aggregation/balance.py:13-185).

Accept neighbor j iff  ||x_j - x_own|| <= gamma * exp(-kappa * t/T) * ||x_own||;
if fewer than min_neighbors accepted, accept the closest neighbor; final state
alpha * own + (1 - alpha) * mean(accepted).

MI355X path: distances + norm come from one fused launch pair over [k, P]
(K2/K12), the accept mask and fallback are branchless device-side tensors, and
the final blend is one weighted-sum launch — the whole aggregation is O(1)
kernel launches and zero host syncs (the reference does k+1 ``.item()``-laden
distance computations plus Python averaging, balance.py:82-175).
"""

from __future__ import annotations

import math
from typing import Any, Dict, List

from torch import Tensor

from murmura_amd import ops
from murmura_amd.aggregation.base import Aggregator, _to_float_list, accept_weights, blend


class BALANCEAggregator(Aggregator):
    def __init__(
        self,
        gamma: float = 2.0,
        kappa: float = 1.0,
        alpha: float = 0.5,
        min_neighbors: int = 1,
        total_rounds: int = 50,
    ):
        self.gamma = float(gamma)
        self.kappa = float(kappa)
        self.alpha = float(alpha)
        self.min_neighbors = int(min_neighbors)
        self.total_rounds = int(total_rounds)
        self._acceptance_history: List[Tensor] = []
        self._threshold_history: List[Tensor] = []

    def _decay(self, round_num: int) -> float:
        t_frac = round_num / max(1, self.total_rounds)
        return self.gamma * math.exp(-self.kappa * t_frac)

    def aggregate(
        self,
        node_id: int,
        own_state: Tensor,
        neighbor_states: Tensor,
        round_num: int = 0,
        pairwise_d2: Tensor = None,
        **ctx: Any,
    ) -> Tensor:
        """``pairwise_d2``: optional [1+k, 1+k] squared distances over
        [own] + neighbors (chunked-exchange path); row 0 gives the distances
        to own and entry (0, 0)... the own norm comes via the caller's Gram
        diagonal, so we derive both from it when present."""
        k = neighbor_states.shape[0]
        if k == 0:
            return own_state.clone()
        if pairwise_d2 is not None:
            dists = pairwise_d2[0, 1:].clamp_min(0).sqrt()
            own_norm = ctx.get("own_norm")
            if own_norm is None:
                own_norm = ops.row_norms(own_state.unsqueeze(0)).squeeze(0)
        else:
            own_norm = ops.row_norms(own_state.unsqueeze(0)).squeeze(0)
            dists = ops.l2_dists_to(own_state, neighbor_states)
        threshold = self._decay(round_num) * own_norm
        accept = dists <= threshold
        w = accept_weights(accept, dists, self.min_neighbors)
        self._acceptance_history.append(accept.float().mean())
        self._threshold_history.append(threshold)
        return blend(own_state, neighbor_states, w, self.alpha)

    def get_statistics(self) -> Dict[str, Any]:
        return {
            "acceptance_rates": _to_float_list(self._acceptance_history),
            "thresholds": _to_float_list(self._threshold_history),
        }
