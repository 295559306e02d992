"""Mobility-driven dynamic topology G^t (reference: murmura/topology/dynamic.py:16-105).

Bounded random walk on a 2-D torus. Fully deterministic from the seed: every
process computes identical positions and therefore identical G^t per round with
no communication — this is the property that lets the RCCL backend agree on the
per-round P2P exchange plan without any coordination messages.
"""

from __future__ import annotations

from typing import Dict, Set, Tuple

import numpy as np

from murmura_amd.topology.base import Topology


class MobilityModel:
    def __init__(
        self,
        num_nodes: int,
        area_size: float = 100.0,
        comm_range: float = 30.0,
        max_speed: float = 5.0,
        seed: int = 42,
        ensure_connected: bool = True,
    ) -> None:
        self.num_nodes = num_nodes
        self.area_size = float(area_size)
        self.comm_range = float(comm_range)
        self.max_speed = float(max_speed)
        self.seed = seed
        self.ensure_connected = ensure_connected
        rng = np.random.default_rng(seed)
        self._initial = rng.uniform(0.0, self.area_size, size=(num_nodes, 2))
        self._rng = rng
        # positions memoized per round; round r depends on rounds 0..r-1 having
        # been generated (walk is sequential), so we generate lazily in order.
        self._positions: Dict[int, np.ndarray] = {0: self._initial.copy()}
        self._max_generated = 0

    def positions_at(self, round_num: int) -> np.ndarray:
        """Positions [num_nodes, 2] at a given round (memoized, generated in order)."""
        if round_num < 0:
            raise ValueError("round_num must be >= 0")
        while self._max_generated < round_num:
            prev = self._positions[self._max_generated]
            delta = self._rng.uniform(
                -self.max_speed, self.max_speed, size=(self.num_nodes, 2)
            )
            nxt = np.mod(prev + delta, self.area_size)
            self._max_generated += 1
            self._positions[self._max_generated] = nxt
        return self._positions[round_num]

    def _torus_dist(self, a: np.ndarray, b: np.ndarray) -> float:
        d = np.abs(a - b)
        d = np.minimum(d, self.area_size - d)
        return float(np.sqrt(np.sum(d * d)))

    def topology_at(self, round_num: int) -> Topology:
        """G^t: edge (i,j) iff torus-distance < comm_range; optionally attach
        isolated nodes to their nearest peer."""
        pos = self.positions_at(round_num)
        n = self.num_nodes
        edges: Set[Tuple[int, int]] = set()
        for i in range(n):
            for j in range(i + 1, n):
                if self._torus_dist(pos[i], pos[j]) < self.comm_range:
                    edges.add((i, j))
        if self.ensure_connected and n > 1:
            deg = {i: 0 for i in range(n)}
            for i, j in edges:
                deg[i] += 1
                deg[j] += 1
            for i in range(n):
                if deg[i] == 0:
                    # attach to nearest peer
                    dists = [
                        (self._torus_dist(pos[i], pos[j]), j) for j in range(n) if j != i
                    ]
                    _, j = min(dists)
                    edges.add((min(i, j), max(i, j)))
                    deg[i] += 1
                    deg[j] += 1
        return Topology.from_edges(n, edges)
