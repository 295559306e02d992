"""Static topology generators (reference: murmura/topology/generators.py:11-140).

All generators are deterministic from their arguments so every rank of the RCCL
backend can construct an identical Topology with no communication.
"""

from __future__ import annotations

import random
import warnings
from typing import Optional, Set, Tuple

from murmura_amd.topology.base import Topology


def create_topology(
    topology_type: str,
    num_nodes: int,
    p: float = 0.3,
    k: int = 4,
    seed: Optional[int] = None,
) -> Topology:
    """Dispatch on topology type string.

    Accepted aliases follow the reference: ring | fully/full |
    erdos/er/erdos-renyi | k-regular/kregular.
    """
    if num_nodes < 1:
        raise ValueError(f"num_nodes must be >= 1, got {num_nodes}")
    t = topology_type.lower()
    if t == "ring":
        return _create_ring(num_nodes)
    if t in ("fully", "full"):
        return _create_fully_connected(num_nodes)
    if t in ("erdos", "er", "erdos-renyi"):
        return _create_erdos_renyi(num_nodes, p=p, seed=seed)
    if t in ("k-regular", "kregular"):
        return _create_k_regular(num_nodes, k=k)
    raise ValueError(f"unknown topology type: {topology_type!r}")


def _create_ring(n: int) -> Topology:
    edges: Set[Tuple[int, int]] = set()
    if n == 2:
        edges.add((0, 1))
    elif n > 2:
        for i in range(n):
            edges.add((min(i, (i + 1) % n), max(i, (i + 1) % n)))
    return Topology.from_edges(n, edges)


def _create_fully_connected(n: int) -> Topology:
    edges = {(i, j) for i in range(n) for j in range(i + 1, n)}
    return Topology.from_edges(n, edges)


def _create_erdos_renyi(n: int, p: float, seed: Optional[int]) -> Topology:
    """Seeded p-coin per pair; isolated node i is connected to (i+1)%n to keep
    the graph usable (reference: generators.py:97-103)."""
    rng = random.Random(seed)
    edges: Set[Tuple[int, int]] = set()
    for i in range(n):
        for j in range(i + 1, n):
            if rng.random() < p:
                edges.add((i, j))
    # connect isolated nodes
    if n > 1:
        deg = {i: 0 for i in range(n)}
        for i, j in edges:
            deg[i] += 1
            deg[j] += 1
        for i in range(n):
            if deg[i] == 0:
                j = (i + 1) % n
                edges.add((min(i, j), max(i, j)))
                deg[i] += 1
                deg[j] += 1
    return Topology.from_edges(n, edges)


def _create_k_regular(n: int, k: int) -> Topology:
    """Circulant lattice with k/2 offsets each side; odd k is bumped to k+1
    with a warning; k >= n falls back to fully connected
    (reference: generators.py:111-140)."""
    if k >= n:
        warnings.warn(
            f"k={k} >= num_nodes={n}; falling back to fully connected", stacklevel=2
        )
        return _create_fully_connected(n)
    if k % 2 == 1:
        warnings.warn(f"k must be even for circulant k-regular; using k={k + 1}", stacklevel=2)
        k = k + 1
        if k >= n:
            return _create_fully_connected(n)
    edges: Set[Tuple[int, int]] = set()
    for i in range(n):
        for off in range(1, k // 2 + 1):
            j = (i + off) % n
            edges.add((min(i, j), max(i, j)))
    return Topology.from_edges(n, edges)
