from murmura_amd.models.zoo import (
    MODEL_FACTORIES,
    BasicBlock,
    CelebAModel,
    FEMNISTModel,
    ResNet18,
    SimpleMLP,
    count_params,
    get_model_variant,
)
from murmura_amd.models.evidential import (
    EvidentialHARClassifier,
    EvidentialHead,
    EvidentialLoss,
    EvidentialPAMAP2Classifier,
    EvidentialPPGDaLiAClassifier,
    compute_uncertainty,
    get_evidential_loss,
)

__all__ = [
    "SimpleMLP",
    "FEMNISTModel",
    "CelebAModel",
    "ResNet18",
    "BasicBlock",
    "get_model_variant",
    "count_params",
    "MODEL_FACTORIES",
    "EvidentialHead",
    "EvidentialLoss",
    "EvidentialHARClassifier",
    "EvidentialPAMAP2Classifier",
    "EvidentialPPGDaLiAClassifier",
    "compute_uncertainty",
    "get_evidential_loss",
]
