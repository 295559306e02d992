"""UBAR: two-stage Byzantine-resilient aggregation
(reference: murmura/aggregation/ubar.py:15-271).

Stage 1 (distance): keep the rho * |N| closest neighbors by full-state L2.
Stage 2 (performance): evaluate each candidate state's CE loss on ONE local
training batch; keep candidates whose loss <= own loss, falling back to the
single best-loss candidate. Final state: alpha-blend as BALANCE.

MI355X path: distances are one fused launch (K2); candidate evaluation swaps
each candidate flat vector into a scratch model bound to a flat buffer (a
device-to-device copy, not the reference's deepcopy + load_state_dict per
neighbor); the keep mask, fallback and blend are branchless device tensors.
"""

from __future__ import annotations

from typing import Any, Dict, List

import torch
from torch import Tensor

from murmura_amd import ops
from murmura_amd.aggregation.base import (
    Aggregator,
    EvalContext,
    _to_float_list,
    blend,
)


class UBARAggregator(Aggregator):
    requires_eval_context = True

    def __init__(
        self,
        rho: float = 0.4,
        alpha: float = 0.5,
        min_neighbors: int = 1,
        total_rounds: int = 50,
    ):
        self.rho = float(rho)
        self.alpha = float(alpha)
        self.min_neighbors = int(min_neighbors)
        self.total_rounds = int(total_rounds)
        self._stage1_kept: List[float] = []
        self._stage2_kept: List[Tensor] = []

    def aggregate(
        self,
        node_id: int,
        own_state: Tensor,
        neighbor_states: Tensor,
        round_num: int = 0,
        eval_context: EvalContext = None,
        **ctx: Any,
    ) -> Tensor:
        k = neighbor_states.shape[0]
        if k == 0:
            return own_state.clone()
        if eval_context is None:
            # reference falls back to distance-only behavior when no eval
            # context is provided (ubar.py:56-99 requires kwargs)
            raise ValueError("UBAR requires an eval_context (train batch + model template)")

        # ---- stage 1: distance filter (device-side); floor on rho*k
        # matches the reference (ubar.py:135 int(self.rho * num_neighbors))
        dists = ops.l2_dists_to(own_state, neighbor_states)
        num_keep = max(self.min_neighbors, int(self.rho * k))
        num_keep = min(num_keep, k)
        _, keep_idx = torch.topk(dists, num_keep, largest=False)
        candidates = neighbor_states.index_select(0, keep_idx)

        # ---- stage 2: performance filter on one training batch — own +
        # all candidates scored in ONE vmapped forward (no per-candidate
        # model swap; reference looped deepcopy+forward, ubar.py:152-222)
        batch = eval_context.next_batch()
        all_losses = eval_context.losses_on_batch(
            torch.cat([own_state.unsqueeze(0), candidates], dim=0), batch
        )
        own_loss, cand_losses = all_losses[0], all_losses[1:]
        keep2 = cand_losses <= own_loss
        cnt = keep2.sum()
        fallback = torch.zeros_like(cand_losses)
        fallback[torch.argmin(cand_losses)] = 1.0
        w_cand = keep2.float() / cnt.clamp(min=1).float()
        use_fb = (cnt < 1).float()
        w_cand = use_fb * fallback + (1.0 - use_fb) * w_cand

        self._stage1_kept.append(float(num_keep))
        self._stage2_kept.append(cnt.float())
        return blend(own_state, candidates, w_cand, self.alpha)

    def get_statistics(self) -> Dict[str, Any]:
        return {
            "stage1_kept": self._stage1_kept,
            "stage2_kept": _to_float_list(self._stage2_kept),
            "rho": self.rho,
        }
