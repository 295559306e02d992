"""Krum (Multi-Krum select-1) — reference: murmura/aggregation/krum.py:8-75.

Mechanism: full m x m pairwise L2 distance matrix over (own + neighbors);
score_i = sum of the (m - c - 2) smallest distances to others; the state with
the minimum score is returned verbatim. Constraint c < (m - 2) / 2, else fall
back to own state (krum.py:49-52).

MI355X path: the distance matrix is ONE fused Gram-matrix kernel over [m, P]
(each row read once — K2 in SURVEY.md §2.9) instead of the reference's
m^2 x num_keys Python loop with per-tensor ``.item()`` syncs; selection stays
on-device (index_select by a 0-dim device tensor, no host round trip).
"""

from __future__ import annotations

from typing import Any, Dict

import torch
from torch import Tensor

from murmura_amd import ops
from murmura_amd.aggregation.base import Aggregator


class KrumAggregator(Aggregator):
    def __init__(self, num_compromised: int = 0):
        self.num_compromised = int(num_compromised)
        self._fallbacks = 0
        self._selections = 0

    def aggregate(
        self,
        node_id: int,
        own_state: Tensor,
        neighbor_states: Tensor,
        round_num: int = 0,
        pairwise_d2: Tensor = None,
        **ctx: Any,
    ) -> Tensor:
        """``pairwise_d2``: optional precomputed [m, m] squared-distance
        matrix over [own] + neighbors in order — supplied by the chunked
        exchange path, which accumulates the Gram matrix while state chunks
        are still on the xGMI wire."""
        m = 1 + neighbor_states.shape[0]
        if self.num_compromised >= (m - 2) / 2 or m < 3:
            self._fallbacks += 1
            return own_state.clone()
        stacked = torch.cat([own_state.unsqueeze(0), neighbor_states], dim=0)
        d2 = pairwise_d2 if pairwise_d2 is not None else ops.pairwise_sq_dists(stacked)
        idx = ops.krum_select(d2, self.num_compromised)
        self._selections += 1
        return stacked.index_select(0, idx.view(1)).squeeze(0).clone()

    def get_statistics(self) -> Dict[str, Any]:
        return {"selections": self._selections, "fallbacks": self._fallbacks}
