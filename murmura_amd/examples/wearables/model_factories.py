"""String -> model factory for evidential wearable classifiers
(reference: murmura/examples/wearables/models.py:355-481)."""

from __future__ import annotations

from typing import Callable

from torch import nn

from murmura_amd.models.evidential import (
    EvidentialHARClassifier,
    EvidentialPAMAP2Classifier,
    EvidentialPPGDaLiAClassifier,
)

_NAMES = {
    # reference factory names (reference: models.py:447-454)
    "uci_har": EvidentialHARClassifier,
    "pamap2": EvidentialPAMAP2Classifier,
    "ppg_dalia": EvidentialPPGDaLiAClassifier,
    "har_classifier": EvidentialHARClassifier,
    "evidential_har": EvidentialHARClassifier,
    "pamap2_classifier": EvidentialPAMAP2Classifier,
    "evidential_pamap2": EvidentialPAMAP2Classifier,
    "ppg_dalia_classifier": EvidentialPPGDaLiAClassifier,
    "evidential_ppg_dalia": EvidentialPPGDaLiAClassifier,
}


def get_factory(name: str, **params) -> Callable[[], nn.Module]:
    key = name.lower().replace("-", "_")
    if key not in _NAMES:
        raise ValueError(f"unknown wearables model {name!r}; options: {sorted(set(_NAMES))}")
    cls = _NAMES[key]
    params.pop("evidential", None)  # marker flag, not a ctor arg
    return lambda: cls(**params)
