"""Topology dataclass (reference: murmura/topology/base.py:8-61)."""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Set, Tuple


@dataclass
class Topology:
    """An undirected communication graph over FL nodes.

    neighbors: adjacency list, neighbors[i] = sorted list of node ids adjacent to i.
    edges: undirected edge set as (i, j) with i < j.
    """

    num_nodes: int
    neighbors: Dict[int, List[int]] = field(default_factory=dict)
    edges: List[Tuple[int, int]] = field(default_factory=list)

    def __post_init__(self) -> None:
        for i in range(self.num_nodes):
            self.neighbors.setdefault(i, [])

    @classmethod
    def from_edges(cls, num_nodes: int, edges: Set[Tuple[int, int]]) -> "Topology":
        norm = sorted({(min(i, j), max(i, j)) for i, j in edges if i != j})
        nbrs: Dict[int, List[int]] = {i: [] for i in range(num_nodes)}
        for i, j in norm:
            nbrs[i].append(j)
            nbrs[j].append(i)
        for i in nbrs:
            nbrs[i].sort()
        return cls(num_nodes=num_nodes, neighbors=nbrs, edges=norm)

    def degree(self, node: int) -> int:
        return len(self.neighbors[node])

    def avg_degree(self) -> float:
        if self.num_nodes == 0:
            return 0.0
        return sum(len(v) for v in self.neighbors.values()) / self.num_nodes

    def is_connected(self) -> bool:
        """BFS connectivity check."""
        if self.num_nodes == 0:
            return True
        seen = {0}
        frontier = [0]
        while frontier:
            nxt = []
            for u in frontier:
                for v in self.neighbors[u]:
                    if v not in seen:
                        seen.add(v)
                        nxt.append(v)
            frontier = nxt
        return len(seen) == self.num_nodes
