"""Core type aliases and protocols (reference: murmura/core/types.py:8-45)."""

from __future__ import annotations

from typing import Dict, List, Protocol, runtime_checkable

import torch
from torch import Tensor

# A model state as a name -> tensor mapping (reference: types.py:8). The
# MI355X-native representation is the flat vector (core/flat.py); dict form is
# kept for checkpointing and API parity.
ModelState = Dict[str, Tensor]

# Per-client dataset index lists (reference: types.py:11)
DataPartition = List[List[int]]


@runtime_checkable
class ModelProtocol(Protocol):
    """Structural protocol for models (reference: types.py:16)."""

    def forward(self, x: Tensor) -> Tensor: ...

    def state_dict(self) -> Dict[str, Tensor]: ...

    def load_state_dict(self, state: Dict[str, Tensor], strict: bool = True): ...

    def to(self, device: torch.device): ...
