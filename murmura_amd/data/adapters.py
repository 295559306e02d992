"""Dataset adapters: dataset + per-client index partitions
(reference: murmura/data/adapters.py:7-57)."""

from __future__ import annotations

from typing import List

from torch.utils.data import Dataset, Subset


class DatasetAdapter:
    """Wraps a torch Dataset plus client_partitions: List[List[int]]; each
    client sees a Subset of the underlying dataset."""

    def __init__(self, dataset: Dataset, client_partitions: List[List[int]]) -> None:
        self.dataset = dataset
        self.client_partitions = [list(p) for p in client_partitions]

    def get_client_data(self, client_id: int) -> Dataset:
        if not (0 <= client_id < len(self.client_partitions)):
            raise IndexError(
                f"client_id {client_id} out of range [0, {len(self.client_partitions)})"
            )
        return Subset(self.dataset, self.client_partitions[client_id])

    def get_num_clients(self) -> int:
        return len(self.client_partitions)

    def get_client_partitions(self) -> List[List[int]]:
        return self.client_partitions


# Alias kept for API familiarity with the reference (adapters.py:55)
TorchDatasetAdapter = DatasetAdapter
