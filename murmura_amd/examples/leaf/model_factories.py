"""String -> model factory for LEAF models
(reference: murmura/examples/leaf/model_factories.py:9-31)."""

from __future__ import annotations

from typing import Callable

from torch import nn

from murmura_amd.models.zoo import CelebAModel, FEMNISTModel, get_model_variant

_NAMES = {
    "femnist": lambda **kw: FEMNISTModel(**kw),
    "femnist_tiny": lambda **kw: get_model_variant("tiny", **kw),
    "femnist_small": lambda **kw: get_model_variant("small", **kw),
    "femnist_baseline": lambda **kw: get_model_variant("baseline", **kw),
    "femnist_large": lambda **kw: get_model_variant("large", **kw),
    "femnist_xlarge": lambda **kw: get_model_variant("xlarge", **kw),
    "celeba": lambda **kw: CelebAModel(**kw),
}


def get_factory(name: str, **params) -> Callable[[], nn.Module]:
    key = name.lower()
    if key not in _NAMES:
        raise ValueError(f"unknown LEAF model {name!r}; options: {sorted(_NAMES)}")
    return lambda: _NAMES[key](**params)
