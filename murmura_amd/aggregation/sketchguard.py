"""Sketchguard: Count-Sketch-compressed BALANCE filtering
(reference: murmura/aggregation/sketchguard.py:13-274).

The filtering decision runs on sketch_size-dim Count-Sketch projections
(s[h[p]] += sign[p] * x[p]) instead of full P-vectors; aggregation of accepted
neighbors uses the full states (alpha-blend, identical to BALANCE). An
adaptive attack factor widens the threshold by 1.5x when the mean of the last
``attack_detection_window``-windowed acceptance rates drops below 0.3
(sketchguard.py:189-204).

MI355X path: sketches for all m states come from ONE LDS-privatized histogram
kernel over [m, P] (K4); hash/sign tables are deterministic from network_seed
so every rank builds identical tables with no communication. The RCCL backend
can exchange 4 KB sketches instead of full states for the filtering pass
(``sketch_wire_mode`` — the comm-saving mode the reference left latent,
sketchguard.py:114-132): pass precomputed ``neighbor_sketches`` via ctx.
"""

from __future__ import annotations

import math
from typing import Any, Dict, List, Optional

import torch
from torch import Tensor

from murmura_amd import ops
from murmura_amd.aggregation.base import Aggregator, _to_float_list, accept_weights, blend


class SketchguardAggregator(Aggregator):
    def __init__(
        self,
        model_dim: int,
        sketch_size: int = 1000,
        gamma: float = 2.0,
        kappa: float = 1.0,
        alpha: float = 0.5,
        min_neighbors: int = 1,
        network_seed: int = 42,
        attack_detection_window: int = 5,
        total_rounds: int = 50,
    ):
        self.model_dim = int(model_dim)
        self.sketch_size = int(sketch_size)
        self.gamma = float(gamma)
        self.kappa = float(kappa)
        self.alpha = float(alpha)
        self.min_neighbors = int(min_neighbors)
        self.network_seed = int(network_seed)
        self.attack_detection_window = int(attack_detection_window)
        self.total_rounds = int(total_rounds)
        self._tables = None  # (hash_idx, signs) lazily placed on device
        self._acceptance_history: List[Tensor] = []

    def _get_tables(self, device: torch.device):
        if self._tables is None or self._tables[0].device != device:
            self._tables = ops.make_sketch_tables(
                self.model_dim, self.sketch_size, self.network_seed, device
            )
        return self._tables

    def get_sketch(self, flat_state: Tensor) -> Tensor:
        """Public sketch computation for wire-sharing
        (reference: sketchguard.py:114-124)."""
        h, s = self._get_tables(flat_state.device)
        return ops.count_sketch(flat_state, h, s, self.sketch_size)

    def wire_filter(
        self, own_sketch: Tensor, neighbor_sketches: Tensor, round_num: int
    ) -> Tensor:
        """The filtering decision from sketches alone (no full states) — the
        basis of sketch-first wire exchange: returns the boolean accept mask
        over neighbors. Uses the same threshold math as ``aggregate``."""
        dists = (neighbor_sketches.float() - own_sketch.float().unsqueeze(0)).norm(dim=1)
        t_frac = round_num / max(1, self.total_rounds)
        threshold = (
            self.gamma
            * math.exp(-self.kappa * t_frac)
            * self._attack_factor(own_sketch.device)
            * own_sketch.float().norm()
        )
        return dists <= threshold

    def _attack_factor(self, device: torch.device) -> Tensor:
        """1.5 if the mean of the last 3 acceptance rates < 0.3 else 1.0,
        as a device scalar (no host sync)."""
        recent = self._acceptance_history[-3:]
        if len(recent) < 3:
            return torch.ones((), device=device)
        mean3 = torch.stack([r.to(device) for r in recent]).mean()
        return torch.where(
            mean3 < 0.3,
            torch.full((), 1.5, device=device),
            torch.ones((), device=device),
        )

    def aggregate(
        self,
        node_id: int,
        own_state: Tensor,
        neighbor_states: Tensor,
        round_num: int = 0,
        neighbor_sketches: Optional[Tensor] = None,
        full_neighbor_count: Optional[int] = None,
        **ctx: Any,
    ) -> Tensor:
        """``full_neighbor_count``: in sketch-wire mode the caller prefilters
        and passes only the accepted subset; the acceptance-rate history (which
        drives the adaptive attack factor) must still be measured against the
        FULL neighbor count for parity with the unfiltered path."""
        k = neighbor_states.shape[0]
        if k == 0:
            return own_state.clone()
        h, s = self._get_tables(own_state.device)
        own_sketch = ops.count_sketch(own_state, h, s, self.sketch_size)
        if neighbor_sketches is None:
            neighbor_sketches = ops.count_sketch(neighbor_states, h, s, self.sketch_size)
        own_norm = neighbor_sketches.new_tensor(0.0) + own_sketch.norm()
        dists = (neighbor_sketches.float() - own_sketch.float().unsqueeze(0)).norm(dim=1)
        t_frac = round_num / max(1, self.total_rounds)
        threshold = (
            self.gamma
            * math.exp(-self.kappa * t_frac)
            * self._attack_factor(own_state.device)
            * own_norm
        )
        accept = dists <= threshold
        w = accept_weights(accept, dists, self.min_neighbors)
        if full_neighbor_count is not None and full_neighbor_count > 0:
            self._acceptance_history.append(accept.sum().float() / full_neighbor_count)
        else:
            self._acceptance_history.append(accept.float().mean())
        return blend(own_state, neighbor_states, w, self.alpha)

    def get_statistics(self) -> Dict[str, Any]:
        return {
            "sketch_size": self.sketch_size,
            "model_dim": self.model_dim,
            "compression_ratio": self.model_dim / max(1, self.sketch_size),
            "acceptance_rates": _to_float_list(self._acceptance_history),
        }
