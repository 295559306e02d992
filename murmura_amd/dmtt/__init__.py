from murmura_amd.dmtt.state import DMTTNodeState

__all__ = ["DMTTNodeState"]
