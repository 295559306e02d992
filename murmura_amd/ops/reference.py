"""Torch reference implementations of every hot op (the bit-level oracle).

Each function here is the semantic definition of a HIP kernel in
murmura_amd/ops/hip/ (kernel inventory: SURVEY.md §2.9 K1-K12). The GPU path
must match these within float tolerance; numerics tests compare the HIP kernel
against the fp32 torch implementation of the same op.

All ops operate on flat state vectors [P] or stacked states [m, P]
(see core/flat.py for why).
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch
from torch import Tensor


# ---------------------------------------------------------------- K1
def weighted_sum(stacked: Tensor, weights: Tensor, out: Optional[Tensor] = None) -> Tensor:
    """out[p] = sum_i weights[i] * stacked[i, p].

    Reference semantics: ``average_states`` (murmura/aggregation/base.py:76-115)
    and the alpha-blends of balance/ubar/sketchguard/evidential_trust.
    """
    w = weights.to(stacked.dtype)
    res = torch.mv(stacked.t(), w) if stacked.dim() == 2 else stacked * w
    if out is not None:
        out.copy_(res)
        return out
    return res


# ---------------------------------------------------------------- K2 (+K12)
def pairwise_sq_dists(stacked: Tensor) -> Tensor:
    """D2[i, j] = ||x_i - x_j||^2 over rows, computed via the Gram matrix so
    each row is read once (reference computes per-pair per-key with .item()
    syncs — murmura/aggregation/base.py:118-135; krum.py:54-62)."""
    x = stacked.float()
    g = x @ x.t()
    sq = g.diagonal()
    d2 = sq.unsqueeze(0) + sq.unsqueeze(1) - 2.0 * g
    return d2.clamp_min_(0.0)


def pairwise_l2(stacked: Tensor) -> Tensor:
    """D[i, j] = ||x_i - x_j||_2."""
    return pairwise_sq_dists(stacked).sqrt_()


def row_norms(stacked: Tensor) -> Tensor:
    """||x_i||_2 per row (K12; reference: balance.py:91-97)."""
    return stacked.float().norm(dim=-1)


def l2_dists_to(own: Tensor, stacked: Tensor) -> Tensor:
    """d[i] = ||stacked[i] - own||_2 (BALANCE/UBAR stage-1 filtering)."""
    return (stacked.float() - own.float().unsqueeze(0)).norm(dim=-1)


# ---------------------------------------------------------------- K3
def krum_scores(d2: Tensor, num_compromised: int) -> Tensor:
    """score_i = sum of the (m - c - 2) smallest squared distances to others.

    The reference sorts plain L2 distances and sums the m-c-2 smallest
    (krum.py:64-71); we keep squared distances (monotone => same argmin and
    same ordering of candidate sets as summing monotone transforms? NO —
    sums of squares differ from sums of norms). For exact parity we take
    sqrt first.
    """
    m = d2.shape[0]
    d = d2.clamp_min(0).sqrt()
    # exclude self-distance (diagonal) by setting it to +inf before topk
    dd = d.clone()
    dd.fill_diagonal_(float("inf"))
    k = max(m - num_compromised - 2, 1)
    k = min(k, m - 1)
    smallest, _ = torch.topk(dd, k, dim=1, largest=False)
    return smallest.sum(dim=1)


def krum_select(d2: Tensor, num_compromised: int) -> Tensor:
    """argmin of krum scores, returned as a 0-dim device tensor (no host sync)."""
    return torch.argmin(krum_scores(d2, num_compromised))


# ---------------------------------------------------------------- K4/K5
def count_sketch(stacked: Tensor, hash_idx: Tensor, signs: Tensor, sketch_size: int) -> Tensor:
    """s[i, h[p]] += sign[p] * stacked[i, p] for each row i
    (reference: np.bincount with weights, sketchguard.py:91-112)."""
    x = stacked.float() if stacked.dim() == 2 else stacked.float().unsqueeze(0)
    m = x.shape[0]
    out = torch.zeros(m, sketch_size, device=x.device, dtype=torch.float32)
    idx = hash_idx.long().unsqueeze(0).expand(m, -1)
    out.scatter_add_(1, idx, x * signs.float())
    return out if stacked.dim() == 2 else out.squeeze(0)


SKETCH_GROUP = 8


def make_sketch_tables(
    model_dim: int, sketch_size: int, seed: int, device: torch.device,
    group: int = SKETCH_GROUP,
) -> Tuple[Tensor, Tensor]:
    """Hash/sign tables from a seeded RNG (reference: sketchguard.py:71-76 uses
    np.random.RandomState; we use torch.Generator — deterministic per seed,
    shared across all ranks by construction).

    MI355X-native table structure: the BIN assignment is constant within each
    aligned ``group`` of 8 consecutive coordinates while SIGNS stay
    per-element. Per-element signs keep every cross term zero-mean, so sketch
    distances remain unbiased; grouping only raises the collision variance of
    coordinate PAIRS inside one group (8/S of coordinates instead of 1/S —
    negligible at S=1000). What it buys: the GPU kernel accumulates each
    group in registers and issues ONE LDS atomic per group per row instead of
    8 — the kernel is atomic-issue-bound (measured ~200G atomics/s), so this
    is the difference between 2.4 and >10 TB/s effective. The CPU reference
    path below and this table layout produce IDENTICAL sketch values.
    Pass ``group=1`` for reference-style fully independent bins."""
    g = torch.Generator().manual_seed(seed)
    if group > 1:
        ngroups = (model_dim + group - 1) // group
        hg = torch.randint(0, sketch_size, (ngroups,), generator=g, dtype=torch.int64)
        hash_idx = hg.repeat_interleave(group)[:model_dim]
    else:
        hash_idx = torch.randint(
            0, sketch_size, (model_dim,), generator=g, dtype=torch.int64
        )
    signs = torch.randint(0, 2, (model_dim,), generator=g, dtype=torch.int64) * 2 - 1
    return hash_idx.to(device), signs.to(device=device, dtype=torch.float32)


# ---------------------------------------------------------------- K6
def sgd_step(flat_params: Tensor, grad: Tensor, lr: float) -> None:
    """p <- p - lr * g, fused across the whole param prefix
    (reference: fresh torch.optim.SGD per round, core/node.py:74-98)."""
    flat_params.add_(grad, alpha=-lr)


# ---------------------------------------------------------------- K7
def ce_loss_acc(logits: Tensor, targets: Tensor) -> Tuple[Tensor, Tensor]:
    """(summed CE loss, correct count) — fused eval epilogue
    (reference: utils/metrics.py:9-47, ubar.py:204-222)."""
    logits = logits.float()
    loss = torch.nn.functional.cross_entropy(logits, targets, reduction="sum")
    correct = (logits.argmax(dim=1) == targets).sum()
    return loss, correct


# ---------------------------------------------------------------- K8
def evidential_stats(
    logits: Tensor, targets: Tensor
) -> Tuple[Tensor, Tensor, Tensor, Tensor]:
    """Per-batch sums of (vacuity, entropy, strength, correct) from evidential
    logits: alpha = softplus(logits) + 1, S = sum(alpha), vacuity = K/S,
    entropy over p = alpha/S, strength = S
    (reference: core/node.py:150-179; evidential_trust.py:249-287)."""
    logits = logits.float()
    alpha = torch.nn.functional.softplus(logits) + 1.0
    s = alpha.sum(dim=1)
    k = float(logits.shape[1])
    vacuity = k / s
    p = alpha / s.unsqueeze(1)
    entropy = -(p * p.clamp_min(1e-10).log()).sum(dim=1)
    correct = (alpha.argmax(dim=1) == targets).sum()
    return vacuity.sum(), entropy.sum(), s.sum(), correct


# ---------------------------------------------------------------- K10
def gaussian_inject(flat: Tensor, noise_std: float, seed: int, offset: int = 0) -> Tensor:
    """attacked = flat + N(0, noise_std^2), deterministic from (seed, offset)
    (reference: attacks/gaussian.py:79-90). The HIP kernel uses Philox with the
    same (seed, offset) counter scheme; CPU parity is statistical, not bitwise."""
    g = torch.Generator(device=flat.device).manual_seed(seed + offset)
    noise = torch.randn(flat.shape, generator=g, device=flat.device, dtype=torch.float32)
    return flat + noise.to(flat.dtype) * noise_std


# ---------------------------------------------------------------- K11
def scale_inject(flat: Tensor, lam: float) -> Tensor:
    """attacked = lam * flat (reference: attacks/directed.py:79-89)."""
    return flat * lam
