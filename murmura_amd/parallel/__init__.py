from murmura_amd.parallel import exchange
from murmura_amd.parallel.runner import DistributedRunner

__all__ = ["exchange", "DistributedRunner"]
