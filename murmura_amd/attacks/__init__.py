from murmura_amd.attacks.base import Attack, select_compromised
from murmura_amd.attacks.gaussian import GaussianAttack
from murmura_amd.attacks.directed import DirectedDeviationAttack
from murmura_amd.attacks.topology_liar import TopologyLiarAttack

__all__ = [
    "Attack",
    "select_compromised",
    "GaussianAttack",
    "DirectedDeviationAttack",
    "TopologyLiarAttack",
]
