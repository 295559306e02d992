"""LEAF dataset -> DatasetAdapter (reference: murmura/examples/leaf/adapter.py:19-61)."""

from __future__ import annotations

from typing import Optional

from murmura_amd.data.adapters import DatasetAdapter
from murmura_amd.examples.leaf.datasets import (
    create_leaf_client_partitions,
    load_leaf_dataset,
)


def load_leaf_adapter(
    dataset_type: str,
    data_path: str,
    split: str = "train",
    max_samples: Optional[int] = None,
    num_nodes: int = 10,
    seed: int = 42,
    **kwargs,
) -> DatasetAdapter:
    ds = load_leaf_dataset(
        dataset_type, data_path, split=split, max_samples=max_samples, **kwargs
    )
    parts = create_leaf_client_partitions(ds, num_nodes, seed=seed)
    return DatasetAdapter(ds, parts)
