"""Fused NHWC BatchNorm2d (K14).

torch's channels-last BatchNorm kernels measured ~230 GB/s on MI355X
(37 us per 8 MB channel reduction) and accounted for ~half of a ResNet-18
bs-64 training step; the HIP kernels behind this module are plain two-pass
streaming reductions with per-thread channel ownership (registers, not LDS
atomics) targeting the HBM roofline.

``MurmuraBatchNorm2d`` subclasses ``nn.BatchNorm2d`` — identical parameters,
buffers, state-dict names and CPU/eager behavior; the fused path engages on
ROCm for channels_last inputs with supported channel counts (C % 8 == 0 and
C | 256 or 256 | C, C <= 1024).
"""

from __future__ import annotations

import torch
from torch import Tensor, nn


def _ext():
    from murmura_amd.ops import _load_ext

    return _load_ext()


def _supported_c(c: int) -> bool:
    # pack-per-thread reduction needs C/8 packs <= 256 threads (bf16; fp32 is
    # C/4 <= 256 -> c <= 1024 covers both)
    return c % 8 == 0 and c <= 1024


class _FusedBNTrain(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, running_mean, running_var, momentum, eps,
                relu, res):
        y, mean, invstd = _ext().bn_fwd_train(
            x, weight, bias, running_mean, running_var, momentum, eps, relu, res
        )
        ctx.relu = relu
        ctx.has_res = res is not None
        if relu:
            ctx.save_for_backward(x, weight, mean, invstd, y)
        else:
            ctx.save_for_backward(x, weight, mean, invstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        if ctx.relu:
            x, weight, mean, invstd, y = ctx.saved_tensors
        else:
            x, weight, mean, invstd = ctx.saved_tensors
            y = None
        dy = dy.contiguous(memory_format=torch.channels_last)
        out = _ext().bn_bwd(x, dy, weight, mean, invstd, y, ctx.relu, ctx.has_res)
        dres = out[3] if ctx.has_res else None
        return out[0], out[1], out[2], None, None, None, None, None, dres


class MurmuraBatchNorm2d(nn.BatchNorm2d):
    """Fused NHWC BatchNorm; set ``fuse_relu=True`` (or use MurmuraBNReLU) to
    fold the subsequent ReLU into the normalize pass and its mask into the
    backward (one fewer elementwise kernel in each direction).

    ``forward(x, res=...)`` additionally fuses a ResNet residual add:
    y = [relu](bn(x) + res), with the residual's gradient produced by the
    same backward kernel (the post-add ReLU's mask applies to both branches).
    Removes 2 elementwise kernels per block per direction at bs-64 shapes
    where kernel count, not bandwidth, bounds the step.

    Class-level ``defer_num_batches_tracked``: when True, the per-forward
    ``num_batches_tracked += 1`` device op is skipped (a ~5 us kernel per BN
    per step inside captured graphs); the training loop calls
    ``bump_num_batches_tracked(model, n)`` once per epoch instead.
    """

    fuse_relu: bool = False
    defer_num_batches_tracked: bool = False

    def forward(self, x: Tensor, res: "Tensor | None" = None) -> Tensor:
        use_fused = (
            x.is_cuda
            and x.dim() == 4
            and x.is_contiguous(memory_format=torch.channels_last)
            and _supported_c(x.shape[1])
            and self.track_running_stats
            and self.affine
            and _ext() is not None
        )
        if not use_fused:
            out = super().forward(x)
            if res is not None:
                out = out + res
            return torch.relu(out) if self.fuse_relu else out
        if res is not None and not res.is_contiguous(memory_format=torch.channels_last):
            res = res.contiguous(memory_format=torch.channels_last)
        if self.training:
            if (self.num_batches_tracked is not None
                    and not MurmuraBatchNorm2d.defer_num_batches_tracked):
                self.num_batches_tracked.add_(1)
            return _FusedBNTrain.apply(
                x, self.weight, self.bias, self.running_mean, self.running_var,
                self.momentum if self.momentum is not None else 0.1, self.eps,
                self.fuse_relu, res,
            )
        return _ext().bn_fwd_eval(
            x, self.weight, self.bias, self.running_mean, self.running_var, self.eps,
            self.fuse_relu, res,
        )


class MurmuraBNReLU(MurmuraBatchNorm2d):
    fuse_relu = True


class MurmuraBNAddReLU(MurmuraBatchNorm2d):
    """y = relu(bn(x) + res) — the ResNet block tail in one kernel."""

    fuse_relu = True


def bump_num_batches_tracked(model: nn.Module, n: int) -> None:
    """Host-side replacement for the deferred per-forward increments: call
    once per epoch with the number of batches trained."""
    for m in model.modules():
        if isinstance(m, nn.BatchNorm2d) and m.num_batches_tracked is not None:
            m.num_batches_tracked.add_(n)
