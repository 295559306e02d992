"""Gaussian noise attack (reference: murmura/attacks/gaussian.py:10-90).

attacked = param + N(0, sigma^2) over the flat float state. On GPU the noise
comes from the Philox kernel (K10) with counter (seed, node_id, round_num) —
deterministic per (node, round) on any rank, no generator state to carry.
"""

from __future__ import annotations

from torch import Tensor

from murmura_amd import ops
from murmura_amd.attacks.base import Attack


class GaussianAttack(Attack):
    def __init__(
        self,
        num_nodes: int,
        percentage: float,
        noise_std: float = 10.0,
        seed: int = 42,
    ):
        super().__init__(num_nodes, percentage, seed)
        self.noise_std = float(noise_std)

    def apply_attack(self, node_id: int, flat_state: Tensor, round_num: int) -> Tensor:
        # counter-style offset: unique per (node, round), identical on all ranks
        offset = node_id * 1_000_003 + round_num
        return ops.gaussian_inject(flat_state, self.noise_std, self.seed, offset)
