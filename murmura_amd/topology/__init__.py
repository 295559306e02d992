from murmura_amd.topology.base import Topology
from murmura_amd.topology.generators import create_topology
from murmura_amd.topology.dynamic import MobilityModel

__all__ = ["Topology", "create_topology", "MobilityModel"]
