"""Topology-liar attack for DMTT (reference: murmura/attacks/topology_liar.py:21-102).

Byzantine nodes lie about their G^t neighborhood: the claimed neighbor set is
(true neighbors UNION all other compromised nodes), inflating apparent
connectivity among colluders. Model poisoning is optionally delegated to a
wrapped inner attack (gaussian / directed_deviation).
"""

from __future__ import annotations

from typing import List, Optional

from torch import Tensor

from murmura_amd.attacks.base import Attack


class TopologyLiarAttack(Attack):
    def __init__(
        self,
        num_nodes: int,
        percentage: float,
        seed: int = 42,
        model_attack: Optional[Attack] = None,
    ):
        super().__init__(num_nodes, percentage, seed)
        self.model_attack = model_attack

    def apply_attack(self, node_id: int, flat_state: Tensor, round_num: int) -> Tensor:
        if self.model_attack is not None:
            return self.model_attack.apply_attack(node_id, flat_state, round_num)
        return flat_state.clone()

    def get_false_claims(
        self, node_id: int, true_neighbors: List[int], round_num: int
    ) -> List[int]:
        """Claimed neighbor list = true neighbors + every other compromised
        node (reference: topology_liar.py:78-102)."""
        claims = set(true_neighbors)
        for c in self._compromised:
            if c != node_id:
                claims.add(c)
        return sorted(claims)
