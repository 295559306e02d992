"""Reference-API compatibility helpers operating on state DICTS.

The reference exposes ``average_states`` / ``compute_model_distance`` /
``flatten_model_state`` / ``calculate_model_dimension`` as public helpers over
``Dict[str, Tensor]`` states (murmura/aggregation/base.py:76-170). The
MI355X-native engine works on flat vectors (core/flat.py), but users migrating
from the reference get the same dict-level functions here; they share the same
semantics (float tensors averaged, non-float tensors copied from the first
state; distances/flattening over float tensors only).
"""

from __future__ import annotations

from typing import Dict, List, Optional, Sequence

import torch
from torch import Tensor, nn

from murmura_amd.core.flat import calculate_model_dimension  # noqa: F401 (re-export)

ModelState = Dict[str, Tensor]


def average_states(
    states: Sequence[ModelState], weights: Optional[Sequence[float]] = None
) -> ModelState:
    """Weighted elementwise average: float tensors are zero-initialized and
    accumulated; non-float tensors (e.g. BatchNorm ``num_batches_tracked``)
    are copied from the first state (reference: base.py:76-115)."""
    if not states:
        raise ValueError("average_states: empty state list")
    if weights is None:
        weights = [1.0 / len(states)] * len(states)
    if len(weights) != len(states):
        raise ValueError("weights length must match states length")
    out: ModelState = {}
    first = states[0]
    for key, ref_t in first.items():
        if torch.is_floating_point(ref_t):
            acc = torch.zeros_like(ref_t, dtype=torch.float32)
            for w, st in zip(weights, states):
                acc += float(w) * st[key].float()
            out[key] = acc.to(ref_t.dtype)
        else:
            out[key] = ref_t.clone()
    return out


def compute_model_distance(a: ModelState, b: ModelState) -> float:
    """L2 distance over float tensors: sqrt of summed squared diffs
    (reference: base.py:118-135)."""
    total = 0.0
    for key, ta in a.items():
        if torch.is_floating_point(ta) and key in b:
            d = (ta.float() - b[key].float()).pow(2).sum()
            total += float(d)
    return float(total) ** 0.5


def flatten_model_state(state: ModelState) -> Tensor:
    """Concatenation of all float tensors (reference: base.py:138-152)."""
    parts: List[Tensor] = [
        t.float().reshape(-1) for t in state.values() if torch.is_floating_point(t)
    ]
    if not parts:
        return torch.empty(0)
    return torch.cat(parts)


def get_model_state(model: nn.Module) -> ModelState:
    """CPU clone of a model's state dict (reference: base.py:54-63)."""
    return {k: v.detach().cpu().clone() for k, v in model.state_dict().items()}


def set_model_state(model: nn.Module, state: ModelState) -> None:
    """Load a state dict in-place (reference: base.py:66-73)."""
    model.load_state_dict({k: v for k, v in state.items()}, strict=True)
