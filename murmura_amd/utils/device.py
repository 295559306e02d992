"""Device selection with explicit per-node GPU pinning.

The reference picks one global device (cuda -> mps -> cpu,
murmura/utils/device.py:6-17) and all simulated nodes share it; here each FL
node can be pinned to ``cuda:(node_id % num_gpus)`` — on one 8xMI355X box the
RCCL backend gives each rank its own GPU, and the simulation backend spreads
nodes over whatever GPUs are visible.
"""

from __future__ import annotations

from typing import Optional

import torch


def get_device(explicit: Optional[str] = None, node_id: int = 0) -> torch.device:
    """Resolve a device. ``explicit`` may be "auto", "cpu", "cuda", "cuda:N".

    With "auto"/"cuda" and multiple visible GPUs, node i is pinned to
    ``cuda:(i % num_gpus)``.
    """
    if explicit not in (None, "auto"):
        return torch.device(explicit)
    if torch.cuda.is_available():
        n = torch.cuda.device_count()
        return torch.device(f"cuda:{node_id % max(1, n)}")
    return torch.device("cpu")
