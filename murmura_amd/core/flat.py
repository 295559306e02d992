"""Flat parameter buffers — the K13 design decision.

Every node's float model state lives in ONE contiguous device tensor; the
model's parameters and float buffers are views into it. This is what lets

* every aggregation kernel (weighted sum, pairwise-L2, count-sketch, attack
  injection, fused SGD) operate on a single P-vector with one launch instead
  of the reference's per-dict-key eager loops with ``.item()`` host syncs
  (reference: murmura/aggregation/base.py:76-170), and
* RCCL move a node's whole state GPU-to-GPU as one send with no
  torch.save/pickle serialization (reference: murmura/distributed/messaging.py:59-68).

Layout: trainable parameters first (the fused-SGD prefix), then float buffers
(BatchNorm running stats — the reference averages those too). Non-float
tensors (e.g. BatchNorm ``num_batches_tracked``) are excluded from the flat
vector and kept node-local, matching the reference's ``average_states`` which
copies rather than averages them (reference: aggregation/base.py:104-113).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, Iterator, List, Optional, Tuple

import torch
from torch import Tensor, nn


@dataclass(frozen=True)
class FlatEntry:
    name: str
    offset: int
    numel: int
    shape: Tuple[int, ...]
    is_param: bool


class FlatParamSpec:
    """Name -> (offset, shape) layout of a model's float state."""

    def __init__(self, entries: List[FlatEntry], param_numel: int, total_numel: int):
        self.entries = entries
        self.param_numel = param_numel  # prefix [0, param_numel) = trainable params
        self.total_numel = total_numel
        self._by_name = {e.name: e for e in entries}

    def __getitem__(self, name: str) -> FlatEntry:
        return self._by_name[name]

    def __contains__(self, name: str) -> bool:
        return name in self._by_name

    def __iter__(self) -> Iterator[FlatEntry]:
        return iter(self.entries)

    @classmethod
    def from_model(cls, model: nn.Module) -> "FlatParamSpec":
        entries: List[FlatEntry] = []
        seen: Dict[int, str] = {}  # dedupe tied tensors
        offset = 0
        for name, p in model.named_parameters():
            if id(p) in seen:
                continue
            seen[id(p)] = name
            entries.append(FlatEntry(name, offset, p.numel(), tuple(p.shape), True))
            offset += p.numel()
        param_numel = offset
        for name, b in model.named_buffers():
            if not torch.is_floating_point(b) or id(b) in seen:
                continue
            seen[id(b)] = name
            entries.append(FlatEntry(name, offset, b.numel(), tuple(b.shape), False))
            offset += b.numel()
        return cls(entries, param_numel, offset)


def calculate_model_dimension(model: nn.Module) -> int:
    """Total float dimension P of a model (reference: aggregation/base.py:155-170)."""
    return FlatParamSpec.from_model(model).total_numel


class FlatParamStore:
    """A model bound to a flat buffer: ``flat`` holds all float state, the
    module's parameters/buffers are reshaped views into it, and (after
    ``ensure_grads``) gradients accumulate into the contiguous ``grad_flat``
    so one fused kernel performs the whole SGD step (K6)."""

    def __init__(
        self,
        model: nn.Module,
        device: torch.device,
        dtype: torch.dtype,
        channels_last: bool = False,
    ):
        model = model.to(device=device, dtype=dtype)
        self.model = model
        self.device = torch.device(device)
        self.dtype = dtype
        # NHWC storage for 4-D (conv) params: the flat slice holds the weight
        # in (O, H, W, I) order and the bound param is the permuted view, so
        # MIOpen sees channels_last weights (no internal batched_transpose on
        # CDNA4) while aggregation/RCCL still see one contiguous P-vector.
        # Layout is identical across nodes, so cross-node math is unaffected.
        self.channels_last = channels_last
        self.spec = FlatParamSpec.from_model(model)
        self.flat = torch.empty(self.spec.total_numel, device=device, dtype=dtype)
        self.grad_flat: Optional[Tensor] = None
        self._bind(copy=True)

    def _is_nhwc(self, e: FlatEntry) -> bool:
        return self.channels_last and e.is_param and len(e.shape) == 4

    def _bound_view(self, buf: Tensor, e: FlatEntry) -> Tensor:
        """The NCHW-shaped (possibly channels_last-strided) tensor over the
        flat slice for entry ``e``."""
        sl = buf[e.offset : e.offset + e.numel]
        if self._is_nhwc(e):
            o, i, h, w = e.shape
            return sl.view(o, h, w, i).permute(0, 3, 1, 2)
        return sl.view(e.shape)

    @torch.no_grad()
    def _write_entry(self, buf: Tensor, e: FlatEntry, src: Tensor) -> None:
        sl = buf[e.offset : e.offset + e.numel]
        if self._is_nhwc(e):
            o, i, h, w = e.shape
            sl.view(o, h, w, i).copy_(src.detach().permute(0, 2, 3, 1))
        else:
            sl.view(e.shape).copy_(src.detach())

    def _bind(self, copy: bool) -> None:
        """Point every param/float-buffer at its view of ``self.flat``."""
        params = dict(self.model.named_parameters())
        buffers = dict(self.model.named_buffers())
        for e in self.spec:
            src = params[e.name] if e.is_param else buffers[e.name]
            if copy:
                self._write_entry(self.flat, e, src)
            view = self._bound_view(self.flat, e)
            if e.is_param:
                params[e.name].data = view
            else:
                # re-register buffer so the module holds the view
                mod_path, _, leaf = e.name.rpartition(".")
                mod = self.model.get_submodule(mod_path) if mod_path else self.model
                setattr(mod, leaf, view)

    def ensure_grads(self) -> Tensor:
        """Allocate the flat grad buffer (param prefix only) and point every
        param's ``.grad`` at its slice so autograd accumulates in place.
        Grad views carry the same (possibly channels_last) strides as their
        params — autograd requires grad layout to match."""
        if self.grad_flat is None:
            self.grad_flat = torch.zeros(
                self.spec.param_numel, device=self.device, dtype=self.dtype
            )
        params = dict(self.model.named_parameters())
        for e in self.spec:
            if e.is_param:
                params[e.name].grad = self._bound_view(self.grad_flat, e)
        return self.grad_flat

    def zero_grad(self) -> None:
        if self.grad_flat is not None:
            self.grad_flat.zero_()

    @torch.no_grad()
    def copy_from_flat(self, vec: Tensor) -> None:
        """Replace the whole state with ``vec`` (the post-aggregation apply)."""
        if vec.numel() != self.flat.numel():
            raise ValueError(f"flat size mismatch: {vec.numel()} vs {self.flat.numel()}")
        self.flat.copy_(vec.to(device=self.device, dtype=self.dtype))

    @torch.no_grad()
    def snapshot(self) -> Tensor:
        """Clone of the current flat state (the pre-round snapshot)."""
        return self.flat.clone()

    def to_state_dict(self) -> Dict[str, Tensor]:
        """Full state dict (flat views reshaped + non-float buffers), for
        checkpointing and reference-API parity. Tensors are NCHW-shaped
        (channels_last entries are permuted views over the flat slice)."""
        out: Dict[str, Tensor] = {}
        for e in self.spec:
            out[e.name] = self._bound_view(self.flat, e)
        for name, b in self.model.named_buffers():
            if not torch.is_floating_point(b):
                out[name] = b
        return out

    @torch.no_grad()
    def load_state_dict(self, state: Dict[str, Tensor]) -> None:
        for e in self.spec:
            if e.name in state:
                self._write_entry(
                    self.flat, e, state[e.name].to(device=self.device, dtype=self.dtype)
                )
        for name, b in self.model.named_buffers():
            if not torch.is_floating_point(b) and name in state:
                b.copy_(state[name].to(b.device))


def flatten_state_dict(state: Dict[str, Tensor], spec: FlatParamSpec) -> Tensor:
    """Pack a state dict into a flat vector using ``spec``'s layout
    (reference-parity helper for ``flatten_model_state``, base.py:138-152)."""
    first = state[spec.entries[0].name]
    out = torch.empty(spec.total_numel, device=first.device, dtype=first.dtype)
    for e in spec:
        out[e.offset : e.offset + e.numel] = state[e.name].reshape(-1)
    return out
