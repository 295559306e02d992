from murmura_amd.aggregation.base import Aggregator, EvalContext, accept_weights, blend
from murmura_amd.aggregation.fedavg import FedAvgAggregator
from murmura_amd.aggregation.krum import KrumAggregator
from murmura_amd.aggregation.balance import BALANCEAggregator
from murmura_amd.aggregation.sketchguard import SketchguardAggregator
from murmura_amd.aggregation.ubar import UBARAggregator
from murmura_amd.aggregation.evidential_trust import EvidentialTrustAggregator
from murmura_amd.aggregation.compat import (
    average_states,
    calculate_model_dimension,
    compute_model_distance,
    flatten_model_state,
    get_model_state,
    set_model_state,
)

ALGORITHMS = {
    "fedavg": FedAvgAggregator,
    "krum": KrumAggregator,
    "balance": BALANCEAggregator,
    "sketchguard": SketchguardAggregator,
    "ubar": UBARAggregator,
    "evidential_trust": EvidentialTrustAggregator,
}

__all__ = [
    "Aggregator",
    "EvalContext",
    "accept_weights",
    "blend",
    "FedAvgAggregator",
    "KrumAggregator",
    "BALANCEAggregator",
    "SketchguardAggregator",
    "UBARAggregator",
    "EvidentialTrustAggregator",
    "ALGORITHMS",
    "average_states",
    "compute_model_distance",
    "flatten_model_state",
    "calculate_model_dimension",
    "get_model_state",
    "set_model_state",
]
