"""3x3 stride-1 conv with the hand-written MFMA weight-gradient kernel (K15).

Forward and data-gradient stay on MIOpen (their igemm kernels are adequate
there); the WEIGHT gradient — where MIOpen's atomic-igemm choice costs
~20 us + 2-3 SubTensorOp launches per conv at the flagship's bs-64 shapes —
runs on `conv3x3s1_wrw` (MFMA 32x32x16 bf16, deterministic in-launch
split-image reduction; see ops/hip/murmura_kernels.hip K15).
"""

from __future__ import annotations

import torch
from torch import Tensor, nn


def _ext():
    from murmura_amd.ops import _load_ext

    return _load_ext()


def _wrw_supported(x: Tensor, weight: Tensor) -> bool:
    import os

    # MURMURA_NATIVE_WRW: "1" enables all supported shapes, "w8" only W=8
    # (the shape where the standalone op beats MIOpen: 53.7 vs 61.0 us).
    # Default OFF: inside the captured flagship epoch the full substitution
    # measured 13.40 vs 13.96 rounds/s — the separate dx (conv2d_input) +
    # slab+reduce pipeline costs more than the per-op win recovers
    # (profiles/r02_mfma_wrw.md).
    mode = os.environ.get("MURMURA_NATIVE_WRW", "0")
    if mode == "0":
        return False
    if not (x.is_cuda and x.dtype == torch.bfloat16):
        return False
    if not x.is_contiguous(memory_format=torch.channels_last):
        return False
    n, c, h, w = x.shape
    k = weight.shape[0]
    ok = (c % 64 == 0 and k % 64 == 0 and c <= 512 and k <= 512
          and w in (4, 8, 16, 32) and h % min(h, 8) == 0 and _ext() is not None)
    if mode == "w8":
        ok = ok and w == 8
    return ok


class _Conv3x3Fn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight):
        ctx.save_for_backward(x, weight)
        return torch.nn.functional.conv2d(x, weight, None, 1, 1)

    @staticmethod
    def backward(ctx, dy):
        x, weight = ctx.saved_tensors
        dy = dy.contiguous(memory_format=torch.channels_last)
        dx = None
        if ctx.needs_input_grad[0]:
            # dx through the SAME aten op autograd would use (conv2d_input
            # keys a different MIOpen find entry — measured slower end-to-end)
            dx = torch.ops.aten.convolution_backward(
                dy, x, weight, None, (1, 1), (1, 1), (1, 1), False, (0, 0), 1,
                (True, False, False),
            )[0]
        dw = _ext().conv3x3s1_wrw(x, dy)
        return dx, dw


class MurmuraConv3x3(nn.Conv2d):
    """Drop-in 3x3 s1 p1 bias-free Conv2d whose wgrad uses the MFMA kernel
    when shapes allow; falls back to the plain conv otherwise."""

    def __init__(self, in_ch: int, out_ch: int, stride: int = 1):
        super().__init__(in_ch, out_ch, 3, stride=stride, padding=1, bias=False)

    def forward(self, x: Tensor) -> Tensor:
        if (self.stride == (1, 1) and self.training
                and torch.is_grad_enabled() and _wrw_supported(x, self.weight)):
            return _Conv3x3Fn.apply(x, self.weight)
        return super().forward(x)
