"""Attack protocol (reference: murmura/attacks/base.py:8-52).

Attacks operate on FLAT state vectors on-device (BASELINE.json requires on-GPU
attack injection so the round loop never leaves the device); node selection is
deterministic from the seed so every rank computes the identical compromised
set with no communication.
"""

from __future__ import annotations

import abc
import random
from typing import List

from torch import Tensor


def select_compromised(
    num_nodes: int, percentage: float, seed: int
) -> List[int]:
    """Seeded sample of int(pct * n) nodes (floor — reference:
    attacks/gaussian.py:37-39), at least 1 when pct > 0."""
    if percentage <= 0.0:
        return []
    num = min(num_nodes, max(1, int(percentage * num_nodes)))
    rng = random.Random(seed)
    return sorted(rng.sample(range(num_nodes), num))


class Attack(abc.ABC):
    """Byzantine model-poisoning attack over flat states."""

    def __init__(self, num_nodes: int, percentage: float, seed: int = 42):
        self.num_nodes = num_nodes
        self.percentage = percentage
        self.seed = seed
        self._compromised = select_compromised(num_nodes, percentage, seed)

    def is_compromised(self, node_id: int) -> bool:
        return node_id in self._compromised

    def get_compromised_nodes(self) -> List[int]:
        return list(self._compromised)

    @abc.abstractmethod
    def apply_attack(self, node_id: int, flat_state: Tensor, round_num: int) -> Tensor:
        """Return the attacked flat state (input not mutated)."""
