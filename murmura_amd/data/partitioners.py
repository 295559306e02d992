"""Client index partitioners (reference: murmura/data/partitioners.py:7-223).

All partitioners take a label array and return List[List[int]] of dataset
indices per client, deterministic from the seed.
"""

from __future__ import annotations

from typing import List, Optional, Sequence

import numpy as np


def dirichlet_partition(
    labels: Sequence[int],
    num_clients: int,
    alpha: float = 0.5,
    seed: int = 42,
    min_samples_per_client: int = 2,
) -> List[List[int]]:
    """Non-IID partition: per-class Dirichlet(alpha) proportions over clients
    (reference: partitioners.py:7-124). Small alpha => highly skewed shards.
    Ensures every client ends up with at least ``min_samples_per_client``
    samples by redistributing from the largest clients.
    """
    labels_arr = np.asarray(labels)
    rng = np.random.default_rng(seed)
    classes = np.unique(labels_arr)
    client_indices: List[List[int]] = [[] for _ in range(num_clients)]

    for c in classes:
        idx_c = np.flatnonzero(labels_arr == c)
        rng.shuffle(idx_c)
        props = rng.dirichlet(np.full(num_clients, alpha))
        # integer split with remainder going to the largest-proportion clients
        counts = np.floor(props * len(idx_c)).astype(int)
        remainder = len(idx_c) - int(counts.sum())
        if remainder > 0:
            order = np.argsort(-props)
            for k in range(remainder):
                counts[order[k % num_clients]] += 1
        start = 0
        for client, cnt in enumerate(counts):
            if cnt > 0:
                client_indices[client].extend(idx_c[start : start + cnt].tolist())
            start += cnt

    _ensure_minimum_samples(client_indices, min_samples_per_client, rng)
    for part in client_indices:
        rng.shuffle(part)
    return client_indices


def _ensure_minimum_samples(
    client_indices: List[List[int]], min_samples: int, rng: np.random.Generator
) -> None:
    """Move samples from the largest clients to any client below the minimum
    (reference: partitioners.py:80-124)."""
    for i, part in enumerate(client_indices):
        while len(part) < min_samples:
            donor = max(
                (j for j in range(len(client_indices)) if j != i),
                key=lambda j: len(client_indices[j]),
                default=None,
            )
            if donor is None or len(client_indices[donor]) <= min_samples:
                break
            part.append(client_indices[donor].pop())


def iid_partition(
    num_samples: int, num_clients: int, seed: int = 42
) -> List[List[int]]:
    """Shuffle + even split (reference: partitioners.py:127-150)."""
    rng = np.random.default_rng(seed)
    idx = rng.permutation(num_samples)
    return [a.tolist() for a in np.array_split(idx, num_clients)]


def natural_partition(
    subject_ids: Sequence[int], max_clients: Optional[int] = None
) -> List[List[int]]:
    """Group by subject/user id array; optionally cap to the first N clients
    (reference: partitioners.py:153-181)."""
    sid = np.asarray(subject_ids)
    uniq = np.unique(sid)
    if max_clients is not None:
        uniq = uniq[:max_clients]
    return [np.flatnonzero(sid == u).tolist() for u in uniq]


def combine_partitions_with_dirichlet(
    subject_ids: Sequence[int],
    labels: Sequence[int],
    num_clients: int,
    alpha: float = 0.5,
    seed: int = 42,
) -> List[List[int]]:
    """Re-partition naturally-grouped data with Dirichlet over clients
    (reference: partitioners.py:184-223). Subject grouping is flattened and the
    pooled indices are Dirichlet-partitioned by label."""
    del subject_ids  # pooled: natural grouping is discarded by design
    return dirichlet_partition(labels, num_clients, alpha=alpha, seed=seed)
