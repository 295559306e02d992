"""Wearable dataset -> DatasetAdapter with dirichlet / iid / natural
partitioning (reference: murmura/examples/wearables/adapter.py:18-211)."""

from __future__ import annotations

from typing import Optional

from murmura_amd.data.adapters import DatasetAdapter
from murmura_amd.data.partitioners import (
    dirichlet_partition,
    iid_partition,
    natural_partition,
)
from murmura_amd.examples.wearables.datasets import (
    PAMAP2Dataset,
    PPGDaLiADataset,
    UCIHARDataset,
    get_wearable_dataset_info,
)


def _load_dataset(dataset_type: str, data_path: str, split: str,
                  max_samples: Optional[int]):
    t = dataset_type.lower()
    if t == "uci_har":
        return UCIHARDataset(data_path, split=split, max_samples=max_samples)
    if t == "pamap2":
        return PAMAP2Dataset(data_path, split=split, max_samples=max_samples)
    if t == "ppg_dalia":
        return PPGDaLiADataset(data_path, split=split, max_samples=max_samples)
    raise ValueError(
        f"unknown wearable dataset {dataset_type!r} (uci_har|pamap2|ppg_dalia)"
    )


def load_wearable_adapter(
    dataset_type: str,
    data_path: str,
    num_nodes: int = 10,
    partition_method: str = "dirichlet",
    alpha: float = 0.5,
    seed: int = 42,
    split: str = "train",
    max_samples: Optional[int] = None,
) -> DatasetAdapter:
    ds = _load_dataset(dataset_type, data_path, split, max_samples)
    labels = ds.y.tolist()
    if partition_method == "dirichlet":
        parts = dirichlet_partition(labels, num_nodes, alpha=alpha, seed=seed)
    elif partition_method == "iid":
        parts = iid_partition(len(labels), num_nodes, seed=seed)
    elif partition_method == "natural":
        parts = natural_partition(ds.subjects.tolist(), max_clients=num_nodes)
        # pad with empty clients if fewer subjects than nodes
        while len(parts) < num_nodes:
            parts.append([])
    else:
        raise ValueError(f"unknown partition_method {partition_method!r}")
    return DatasetAdapter(ds, parts)


__all__ = ["load_wearable_adapter", "get_wearable_dataset_info", "_load_dataset"]
