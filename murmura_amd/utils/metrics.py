"""Evaluation helpers (reference: murmura/utils/metrics.py:9-66).

Both evaluators return DEVICE scalars accumulated with the fused eval kernels
(K7/K8); callers batch the host sync once per round instead of per batch.
"""

from __future__ import annotations

from typing import Dict

import torch
from torch import Tensor, nn

from murmura_amd import ops


@torch.no_grad()
def evaluate_model(
    model: nn.Module, loader, device: torch.device, dtype: torch.dtype = torch.float32
) -> Dict[str, Tensor]:
    """Mean CE loss + accuracy over a loader; returns 0-dim device tensors."""
    model.eval()
    loss_sum = torch.zeros((), device=device)
    correct = torch.zeros((), device=device)
    count = 0
    for x, y in loader:
        x = x.to(device=device, dtype=dtype)
        y = y.to(device)
        logits = model(x)
        ls, c = ops.ce_loss_acc(logits, y)
        loss_sum = loss_sum + ls
        correct = correct + c
        count += x.shape[0]
    n = max(1, count)
    return {"loss": loss_sum / n, "accuracy": correct.float() / n, "num_samples": n}


@torch.no_grad()
def evaluate_evidential(
    model: nn.Module, loader, device: torch.device, dtype: torch.dtype = torch.float32
) -> Dict[str, Tensor]:
    """Evidential evaluation: accuracy + mean vacuity/entropy/strength
    (reference: core/node.py:134-196)."""
    model.eval()
    vac = torch.zeros((), device=device)
    ent = torch.zeros((), device=device)
    strength = torch.zeros((), device=device)
    correct = torch.zeros((), device=device)
    loss_sum = torch.zeros((), device=device)
    count = 0
    for x, y in loader:
        x = x.to(device=device, dtype=dtype)
        y = y.to(device)
        logits = model(x)
        v, e, s, c = ops.evidential_stats(logits, y)
        ls, _ = ops.ce_loss_acc(logits, y)
        vac, ent, strength, correct = vac + v, ent + e, strength + s, correct + c
        loss_sum = loss_sum + ls
        count += x.shape[0]
    n = max(1, count)
    return {
        "loss": loss_sum / n,
        "accuracy": correct.float() / n,
        "vacuity": vac / n,
        "entropy": ent / n,
        "strength": strength / n,
        "num_samples": n,
    }


@torch.no_grad()
def compute_accuracy(model: nn.Module, loader, device: torch.device,
                     dtype: torch.dtype = torch.float32) -> float:
    """Plain accuracy over a loader (reference: utils/metrics.py:50-66)."""
    return float(evaluate_model(model, loader, device, dtype)["accuracy"])
