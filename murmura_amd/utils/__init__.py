from murmura_amd.utils.seed import set_seed
from murmura_amd.utils.device import get_device
from murmura_amd.utils.metrics import compute_accuracy, evaluate_evidential, evaluate_model

__all__ = ["set_seed", "get_device", "evaluate_model", "evaluate_evidential", "compute_accuracy"]
