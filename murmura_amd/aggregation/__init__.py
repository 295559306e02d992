from murmura_amd.aggregation.base import Aggregator, EvalContext, accept_weights, blend
from murmura_amd.aggregation.fedavg import FedAvgAggregator
from murmura_amd.aggregation.krum import KrumAggregator
from murmura_amd.aggregation.balance import BALANCEAggregator
from murmura_amd.aggregation.sketchguard import SketchguardAggregator
from murmura_amd.aggregation.ubar import UBARAggregator
from murmura_amd.aggregation.evidential_trust import EvidentialTrustAggregator

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
]
