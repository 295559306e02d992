"""Directed deviation attack (reference: murmura/attacks/directed.py:10-89).

attacked = lambda * param (default lambda = -5.0: sign-flipped and amplified),
one scale kernel over the flat state (K11).
"""

from __future__ import annotations

from torch import Tensor

from murmura_amd import ops
from murmura_amd.attacks.base import Attack


class DirectedDeviationAttack(Attack):
    def __init__(
        self,
        num_nodes: int,
        percentage: float,
        deviation_factor: float = -5.0,
        seed: int = 42,
    ):
        super().__init__(num_nodes, percentage, seed)
        self.deviation_factor = float(deviation_factor)

    def apply_attack(self, node_id: int, flat_state: Tensor, round_num: int) -> Tensor:
        return ops.scale_inject(flat_state, self.deviation_factor)
