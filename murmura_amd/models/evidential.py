"""Evidential deep learning: heads, loss, and the wearables classifiers
(reference: murmura/examples/wearables/models.py:18-347).

Models output raw evidence logits; alpha = softplus(logits) + 1 is taken in
the loss and in the fused evidential-stats kernel (K8), keeping the forward
pass a plain MLP.
"""

from __future__ import annotations

import torch
from torch import Tensor, nn
import torch.nn.functional as F


class EvidentialHead(nn.Module):
    """Final linear layer producing evidence logits
    (reference: models.py:18-46)."""

    def __init__(self, in_features: int, num_classes: int):
        super().__init__()
        self.linear = nn.Linear(in_features, num_classes)
        self.num_classes = num_classes

    def forward(self, x: Tensor) -> Tensor:
        return self.linear(x)


def compute_uncertainty(logits: Tensor) -> dict:
    """Dirichlet uncertainty measures from evidence logits
    (reference: models.py:49-86)."""
    alpha = F.softplus(logits.float()) + 1.0
    s = alpha.sum(dim=1, keepdim=True)
    k = logits.shape[1]
    p = alpha / s
    return {
        "alpha": alpha,
        "strength": s.squeeze(1),
        "vacuity": k / s.squeeze(1),
        "entropy": -(p * p.clamp_min(1e-10).log()).sum(dim=1),
        "probs": p,
    }


class EvidentialLoss(nn.Module):
    """EDL loss: one-hot MSE under Dirichlet expectation + annealed
    KL(Dir(alpha_tilde) || Dir(1)) (reference: models.py:118-179).

    kl_weight ramps linearly to ``max_kl_weight`` over ``annealing_rounds``.
    """

    def __init__(self, num_classes: int, annealing_rounds: int = 10, max_kl_weight: float = 0.1):
        super().__init__()
        self.num_classes = num_classes
        self.annealing_rounds = max(1, int(annealing_rounds))
        self.max_kl_weight = float(max_kl_weight)

    def kl_weight_at(self, round_num: int) -> float:
        return self.max_kl_weight * min(1.0, round_num / self.annealing_rounds)

    def forward(
        self, logits: Tensor, targets: Tensor, round_num: int = 0,
        kl_weight: Tensor = None,
    ) -> Tensor:
        """``kl_weight``: optional 0-dim DEVICE tensor overriding the
        round-derived annealing weight — this is what makes the loss
        hipGraph-capturable (the graph replays with the tensor updated
        per round instead of baking in a Python float)."""
        logits = logits.float()
        alpha = F.softplus(logits) + 1.0
        s = alpha.sum(dim=1, keepdim=True)
        p = alpha / s
        y = F.one_hot(targets, self.num_classes).float()
        # expected MSE under Dirichlet: (y - p)^2 + p(1-p)/(S+1)
        mse = ((y - p) ** 2).sum(dim=1) + (p * (1.0 - p) / (s + 1.0)).sum(dim=1)
        # KL(Dir(alpha_tilde) || Dir(1)) on the misleading evidence
        alpha_t = y + (1.0 - y) * alpha
        kl = self._kl_to_uniform(alpha_t)
        lam = kl_weight if kl_weight is not None else self.kl_weight_at(round_num)
        return (mse + lam * kl).mean()

    def _kl_to_uniform(self, alpha: Tensor) -> Tensor:
        import math

        k = float(self.num_classes)
        s = alpha.sum(dim=1)
        ln_b = torch.lgamma(alpha).sum(dim=1) - torch.lgamma(s)
        # ln B(1,...,1) = -lgamma(K); a plain float keeps this
        # hipGraph-capturable (no H2D copy mid-capture)
        ln_b_uni = math.lgamma(k)
        dg_s = torch.digamma(s).unsqueeze(1)
        term = ((alpha - 1.0) * (torch.digamma(alpha) - dg_s)).sum(dim=1)
        return ln_b_uni - ln_b + term


def _mlp_block(in_f: int, out_f: int, dropout: float = 0.3) -> nn.Sequential:
    return nn.Sequential(
        nn.Linear(in_f, out_f),
        nn.BatchNorm1d(out_f),
        nn.ReLU(),
        nn.Dropout(dropout),
    )


class _EvidentialMLP(nn.Module):
    """MLP body of ``hidden_dims`` blocks + evidential head. Accepts both the
    reference's ctor names (``input_dim``/``hidden_dims``/``dropout``,
    reference: models.py:355-360) and our ``in_features`` alias so verbatim
    reference YAMLs construct models unchanged."""

    def __init__(
        self,
        input_dim: int,
        hidden_dims=(256, 128),
        num_classes: int = 6,
        dropout: float = 0.3,
    ):
        super().__init__()
        dims = [input_dim] + list(hidden_dims)
        self.body = nn.Sequential(
            *[_mlp_block(dims[i], dims[i + 1], dropout) for i in range(len(dims) - 1)]
        )
        self.head = EvidentialHead(dims[-1], num_classes)

    def forward(self, x: Tensor) -> Tensor:
        return self.head(self.body(x.flatten(1)))


def _alias_input_dim(kwargs: dict) -> dict:
    if "in_features" in kwargs:
        kwargs = dict(kwargs)
        kwargs["input_dim"] = kwargs.pop("in_features")
    return kwargs


class EvidentialHARClassifier(_EvidentialMLP):
    """UCI HAR: MLP 561-256-128-EDL6 (reference: models.py:187-243,355-367)."""

    def __init__(self, input_dim: int = 561, hidden_dims=(256, 128),
                 num_classes: int = 6, dropout: float = 0.3, **alias):
        super().__init__(**_alias_input_dim(
            dict(input_dim=input_dim, hidden_dims=hidden_dims,
                 num_classes=num_classes, dropout=dropout, **alias)))


class EvidentialPAMAP2Classifier(_EvidentialMLP):
    """PAMAP2: windows 100x40 -> 4000-512-256-128-EDL12
    (reference: models.py:246-286)."""

    def __init__(self, input_dim: int = 4000, hidden_dims=(512, 256, 128),
                 num_classes: int = 12, dropout: float = 0.3, **alias):
        super().__init__(**_alias_input_dim(
            dict(input_dim=input_dim, hidden_dims=hidden_dims,
                 num_classes=num_classes, dropout=dropout, **alias)))


class EvidentialPPGDaLiAClassifier(_EvidentialMLP):
    """PPG-DaLiA: windows 32x6 -> 192-256-128-64-EDL7
    (reference: models.py:289-347)."""

    def __init__(self, input_dim: int = 192, hidden_dims=(256, 128, 64),
                 num_classes: int = 7, dropout: float = 0.3, **alias):
        super().__init__(**_alias_input_dim(
            dict(input_dim=input_dim, hidden_dims=hidden_dims,
                 num_classes=num_classes, dropout=dropout, **alias)))


def get_evidential_loss(
    num_classes: int, total_rounds: int = 50, max_kl_weight: float = 0.1
) -> EvidentialLoss:
    """Annealing horizon = rounds/2, lambda = 0.1, mirroring the reference's
    build_criterion coupling (utils/factories.py:106-120)."""
    return EvidentialLoss(
        num_classes, annealing_rounds=max(1, total_rounds // 2), max_kl_weight=max_kl_weight
    )
