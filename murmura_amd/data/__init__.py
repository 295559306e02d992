from murmura_amd.data.base import DatasetProtocol
from murmura_amd.data.adapters import DatasetAdapter, TorchDatasetAdapter
from murmura_amd.data.partitioners import (
    combine_partitions_with_dirichlet,
    dirichlet_partition,
    iid_partition,
    natural_partition,
)
from murmura_amd.data.synthetic import (
    load_synthetic_adapter,
    make_synthetic_classification,
    make_synthetic_images,
)

__all__ = [
    "DatasetProtocol",
    "DatasetAdapter",
    "TorchDatasetAdapter",
    "dirichlet_partition",
    "iid_partition",
    "natural_partition",
    "combine_partitions_with_dirichlet",
    "load_synthetic_adapter",
    "make_synthetic_classification",
    "make_synthetic_images",
]
