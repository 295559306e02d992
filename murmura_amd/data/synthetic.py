"""Synthetic datasets for tests and benchmarks (no network access in this
environment — BASELINE.json prescribes synthetic data / random-init weights for
all perf runs)."""

from __future__ import annotations

from typing import List, Optional, Tuple

import torch
from torch.utils.data import TensorDataset

from murmura_amd.data.adapters import DatasetAdapter
from murmura_amd.data.partitioners import dirichlet_partition, iid_partition


def make_synthetic_classification(
    num_samples: int = 1000,
    num_features: int = 20,
    num_classes: int = 4,
    seed: int = 42,
    class_sep: float = 2.0,
) -> TensorDataset:
    """Gaussian blobs around per-class centers — linearly separable enough that
    tiny MLPs converge in a few rounds (used by convergence smoke tests)."""
    g = torch.Generator().manual_seed(seed)
    centers = torch.randn(num_classes, num_features, generator=g) * class_sep
    y = torch.randint(0, num_classes, (num_samples,), generator=g)
    x = centers[y] + torch.randn(num_samples, num_features, generator=g)
    return TensorDataset(x, y)


def make_synthetic_images(
    num_samples: int,
    channels: int = 3,
    height: int = 32,
    width: int = 32,
    num_classes: int = 10,
    seed: int = 42,
) -> TensorDataset:
    """Random images + labels of a given shape (benchmark data)."""
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(num_samples, channels, height, width, generator=g)
    y = torch.randint(0, num_classes, (num_samples,), generator=g)
    return TensorDataset(x, y)


def load_synthetic_adapter(
    num_nodes: int,
    num_samples: int = 1000,
    num_features: int = 20,
    num_classes: int = 4,
    partition: str = "iid",
    alpha: float = 0.5,
    seed: int = 42,
    image_shape: Optional[Tuple[int, int, int]] = None,
) -> DatasetAdapter:
    """Build a DatasetAdapter over synthetic data with iid or dirichlet shards."""
    if image_shape is not None:
        c, h, w = image_shape
        ds = make_synthetic_images(
            num_samples, channels=c, height=h, width=w, num_classes=num_classes, seed=seed
        )
    else:
        ds = make_synthetic_classification(
            num_samples, num_features=num_features, num_classes=num_classes, seed=seed
        )
    labels: List[int] = ds.tensors[1].tolist()
    if partition == "iid":
        parts = iid_partition(len(labels), num_nodes, seed=seed)
    elif partition == "dirichlet":
        parts = dirichlet_partition(labels, num_nodes, alpha=alpha, seed=seed)
    else:
        raise ValueError(f"unknown partition method: {partition!r}")
    return DatasetAdapter(ds, parts)
