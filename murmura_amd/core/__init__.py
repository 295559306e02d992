from murmura_amd.core.types import DataPartition, ModelProtocol, ModelState
from murmura_amd.core.flat import (
    FlatEntry,
    FlatParamSpec,
    FlatParamStore,
    calculate_model_dimension,
    flatten_state_dict,
)
from murmura_amd.core.node import Node
from murmura_amd.core.network import Network, new_history

__all__ = [
    "ModelState",
    "DataPartition",
    "ModelProtocol",
    "FlatEntry",
    "FlatParamSpec",
    "FlatParamStore",
    "calculate_model_dimension",
    "flatten_state_dict",
    "Node",
    "Network",
    "new_history",
]
