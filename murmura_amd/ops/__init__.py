"""Op dispatch: hand-written CDNA4 HIP kernels on GPU, torch reference on CPU.

Policy (per the build rules): when a tensor is on a ROCm device the native
extension MUST be present and is always used — a silent eager fallback on the
GPU box is a bug, so we raise instead. The torch implementations in
``reference.py`` serve the CPU simulation backend and the numerics tests.

Set MURMURA_DISABLE_NATIVE=1 to force the torch path everywhere (debug only).
"""

from __future__ import annotations

import os
from typing import Optional, Tuple

import torch
from torch import Tensor

from murmura_amd.ops import reference as ref

_EXT = None
_EXT_ERR: Optional[str] = None


def _load_ext():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    try:
        from murmura_amd.ops import _hip_loader

        _EXT = _hip_loader.load()
    except Exception as e:  # noqa: BLE001
        _EXT_ERR = f"{type(e).__name__}: {e}"
        _EXT = None
    return _EXT


def native_available() -> bool:
    return _load_ext() is not None


def _use_native(t: Tensor) -> bool:
    if not t.is_cuda:
        return False
    if os.environ.get("MURMURA_DISABLE_NATIVE") == "1":
        return False
    ext = _load_ext()
    if ext is None:
        raise RuntimeError(
            "murmura_amd HIP extension is required for GPU tensors but failed to "
            f"load: {_EXT_ERR}. Build it with `python -m murmura_amd.ops.build` "
            "(or __graft_entry__.build())."
        )
    return True


# ------------------------------------------------------------------ K1
def weighted_sum(stacked: Tensor, weights: Tensor, out: Optional[Tensor] = None) -> Tensor:
    if _use_native(stacked):
        return _EXT.weighted_sum(stacked, weights.to(stacked.device), out)
    return ref.weighted_sum(stacked, weights, out)


# ------------------------------------------------------------------ K2/K12
def pairwise_sq_dists(stacked: Tensor) -> Tensor:
    if _use_native(stacked):
        return _EXT.pairwise_sq_dists(stacked)
    return ref.pairwise_sq_dists(stacked)


def pairwise_l2(stacked: Tensor) -> Tensor:
    return pairwise_sq_dists(stacked).clamp_min(0).sqrt()


def gram(x: Tensor) -> Tensor:
    """Full [m, m] Gram matrix; accepts chunked column views (stride(0) > n)
    so exchange can accumulate it chunk-by-chunk while later chunks are still
    on the wire."""
    if _use_native(x):
        xx = x if x.stride(1) == 1 else x.contiguous()
        return _EXT.gram(xx)
    xf = x.float()
    return xf @ xf.t()


def sq_dists_from_gram(g: Tensor) -> Tensor:
    sq = g.diagonal()
    return (sq.unsqueeze(0) + sq.unsqueeze(1) - 2.0 * g).clamp_min(0.0)


def row_norms(stacked: Tensor) -> Tensor:
    if _use_native(stacked):
        return _EXT.row_norms(stacked.view(1, -1) if stacked.dim() == 1 else stacked).view(
            () if stacked.dim() == 1 else (-1,)
        )
    return ref.row_norms(stacked)


def l2_dists_to(own: Tensor, stacked: Tensor) -> Tensor:
    if _use_native(stacked):
        return _EXT.l2_dists_to(own, stacked)
    return ref.l2_dists_to(own, stacked)


# ------------------------------------------------------------------ K3
def krum_scores(d2: Tensor, num_compromised: int) -> Tensor:
    # m x m is tiny (m <= nodes); torch path is fine on both devices.
    return ref.krum_scores(d2, num_compromised)


def krum_select(d2: Tensor, num_compromised: int) -> Tensor:
    return ref.krum_select(d2, num_compromised)


# ------------------------------------------------------------------ K4/K5
def _packed_sketch_table(hash_idx: Tensor, signs: Tensor, device) -> Tensor:
    """hash+sign packed into one int32 per element (bin in the low 31 bits,
    sign in the sign bit) — halves the kernel's table traffic. Cached on the
    hash tensor (the aggregator reuses its seeded tables every round)."""
    packed = getattr(hash_idx, "_murmura_packed", None)
    if packed is None or packed.device != device:
        neg = (signs < 0).to(torch.int32)
        packed = (hash_idx.to(torch.int32) | (neg << 31)).to(device).contiguous()
        try:
            hash_idx._murmura_packed = packed
        except Exception:
            pass
    return packed


def _grouped_sketch_tables(hash_idx: Tensor, signs: Tensor, device):
    """Detect group-of-8-constant bins (ops.make_sketch_tables' layout) and
    build the packed group table: bin | signbits<<16 per group, plus a
    per-element packed tail for P % 8 leftovers. Cached on the hash tensor."""
    cached = getattr(hash_idx, "_murmura_g8", None)
    if cached is not None and cached[0].device == device:
        return cached
    P = hash_idx.numel()
    ng = P // 8
    if ng == 0 or int(hash_idx.max()) >= (1 << 16):
        return None
    h8 = hash_idx[: ng * 8].view(ng, 8)
    if not bool((h8 == h8[:, :1]).all()):
        return None
    neg = (signs[: ng * 8].view(ng, 8) < 0).to(torch.int32)
    bits = (neg << (16 + torch.arange(8, dtype=torch.int32, device=neg.device))).sum(
        dim=1, dtype=torch.int32
    )
    gt = (h8[:, 0].to(torch.int32) | bits).to(device).contiguous()
    tneg = (signs[ng * 8 :] < 0).to(torch.int32)
    tail = (hash_idx[ng * 8 :].to(torch.int32) | (tneg << 31)).to(device).contiguous()
    out = (gt, tail)
    try:
        hash_idx._murmura_g8 = out
    except Exception:
        pass
    return out


def count_sketch(stacked: Tensor, hash_idx: Tensor, signs: Tensor, sketch_size: int) -> Tensor:
    if _use_native(stacked if stacked.dim() == 2 else stacked.view(1, -1)):
        x = stacked if stacked.dim() == 2 else stacked.view(1, -1)
        if x.shape[0] * sketch_size * 4 <= 64 * 1024 and sketch_size < (1 << 16):
            g8 = _grouped_sketch_tables(hash_idx, signs, x.device)
            if g8 is not None:
                out = _EXT.count_sketch_g8(x, g8[0], g8[1], sketch_size)
                return out if stacked.dim() == 2 else out.view(-1)
        pt = _packed_sketch_table(hash_idx, signs, x.device)
        out = _EXT.count_sketch(x, pt, sketch_size)
        return out if stacked.dim() == 2 else out.view(-1)
    return ref.count_sketch(stacked, hash_idx, signs, sketch_size)


make_sketch_tables = ref.make_sketch_tables


# ------------------------------------------------------------------ K6
def sgd_step(flat_params: Tensor, grad: Tensor, lr: float) -> None:
    if _use_native(flat_params):
        _EXT.sgd_step(flat_params, grad, lr)
        return
    ref.sgd_step(flat_params, grad, lr)


def sgd_step_lrt(flat_params: Tensor, grad: Tensor, lr: Tensor) -> None:
    """K6 with a device-tensor lr (hipGraph-capturable)."""
    if _use_native(flat_params):
        _EXT.sgd_step_lrt(flat_params, grad, lr)
        return
    flat_params.add_(grad * (-lr))


# ------------------------------------------------------------------ K7
def ce_loss_acc(logits: Tensor, targets: Tensor) -> Tuple[Tensor, Tensor]:
    if _use_native(logits):
        out = _EXT.ce_loss_acc(logits, targets)
        return out[0], out[1]
    return ref.ce_loss_acc(logits, targets)


# ------------------------------------------------------------------ K8
def evidential_stats(logits: Tensor, targets: Tensor):
    if _use_native(logits):
        out = _EXT.evidential_stats(logits, targets)
        return out[0], out[1], out[2], out[3]
    return ref.evidential_stats(logits, targets)


# ------------------------------------------------------------------ K10
def gaussian_inject(flat: Tensor, noise_std: float, seed: int, offset: int = 0) -> Tensor:
    if _use_native(flat):
        return _EXT.gaussian_inject(flat, noise_std, seed, offset)
    return ref.gaussian_inject(flat, noise_std, seed, offset)


# ------------------------------------------------------------------ K11
def scale_inject(flat: Tensor, lam: float) -> Tensor:
    if _use_native(flat):
        return _EXT.scale_inject(flat, lam)
    return ref.scale_inject(flat, lam)
