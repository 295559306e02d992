"""murmura_amd — an MI355X-native decentralized federated-learning engine.

A from-scratch rebuild of the capabilities of Cloudslab/murmura
designed for a single 8xMI355X box: one FL node per GPU, model states held in
flat device buffers, neighbor exchange over RCCL/xGMI, and the aggregation hot
paths (weighted averaging, pairwise-L2, Count-Sketch, eval filtering, attack
injection) implemented as hand-written CDNA4 HIP kernels with a CPU/torch
reference path kept as the bit-level oracle.

Public API mirrors the reference's re-exports (reference: murmura/__init__.py:10-19).
"""

from murmura_amd.config.schema import Config
from murmura_amd.config.loader import load_config, save_config
from murmura_amd.core.network import Network
from murmura_amd.core.node import Node
from murmura_amd.core.flat import FlatParamSpec, FlatParamStore
from murmura_amd.topology.base import Topology
from murmura_amd.topology.generators import create_topology
from murmura_amd.topology.dynamic import MobilityModel
from murmura_amd.aggregation.base import Aggregator
from murmura_amd.aggregation.fedavg import FedAvgAggregator
from murmura_amd.aggregation.krum import KrumAggregator
from murmura_amd.aggregation.balance import BALANCEAggregator
from murmura_amd.aggregation.sketchguard import SketchguardAggregator
from murmura_amd.aggregation.ubar import UBARAggregator
from murmura_amd.aggregation.evidential_trust import EvidentialTrustAggregator

__version__ = "0.2.0"

__all__ = [
    "Config",
    "load_config",
    "save_config",
    "Network",
    "Node",
    "FlatParamSpec",
    "FlatParamStore",
    "Topology",
    "create_topology",
    "MobilityModel",
    "Aggregator",
    "FedAvgAggregator",
    "KrumAggregator",
    "BALANCEAggregator",
    "SketchguardAggregator",
    "UBARAggregator",
    "EvidentialTrustAggregator",
]
