"""FedAvg: equal-weight mean of own + neighbor states
(reference: murmura/aggregation/fedavg.py:8-42).

One fused weighted-sum launch over [m, P]. On the fully-connected topology the
RCCL backend folds this into an all-reduce of the flat buffer instead
 — this class is the general-topology path.
"""

from __future__ import annotations

from typing import Any

import torch
from torch import Tensor

from murmura_amd import ops
from murmura_amd.aggregation.base import Aggregator


class FedAvgAggregator(Aggregator):
    def aggregate(
        self,
        node_id: int,
        own_state: Tensor,
        neighbor_states: Tensor,
        round_num: int = 0,
        **ctx: Any,
    ) -> Tensor:
        if neighbor_states.shape[0] == 0:
            return own_state.clone()
        stacked = torch.cat([own_state.unsqueeze(0), neighbor_states], dim=0)
        m = stacked.shape[0]
        w = torch.full((m,), 1.0 / m, device=own_state.device, dtype=torch.float32)
        return ops.weighted_sum(stacked, w)
