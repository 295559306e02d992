"""Device-resident shards + hipGraph-captured round execution.

The FL round loop is launch-bound, not FLOP-bound: the flagship model is a
~11M-param CNN at batch 64, so one eager training step is hundreds of tiny
kernel launches plus a host->device copy per batch. MI355X-native treatment
(per the CDNA4 build rules — HIP graphs instead of a tracing compiler, tensors
resident in 288 GB HBM3E):

- ``DeviceShard``: the node's whole local shard lives on-device once; per-epoch
  shuffling is ONE gather kernel driven by a seeded host permutation (identical
  order to a host sampler with the same generator).
- ``TrainGraph``: the entire local epoch (all batches: zero-grad, forward,
  loss, backward, fused-SGD) is captured into one hipGraph and replayed once
  per round — per-round Python/launch overhead collapses to a shuffle + one
  graph replay.
- ``EvalGraph``: the full evaluation pass (forward + fused CE/evidential stats
  epilogue) is one captured graph as well.

Why replay stays correct across rounds: every parameter and gradient is a view
into the node's FlatParamStore buffer (core/flat.py), and aggregation applies
new states with an in-place copy into that SAME buffer — graph-captured
addresses never go stale.
"""

from __future__ import annotations

from typing import Optional

import torch
from torch import Tensor

from murmura_amd import ops


class DeviceShard:
    """A node's local dataset materialized on its GPU."""

    def __init__(self, x: Tensor, y: Tensor, device: torch.device, dtype: torch.dtype,
                 channels_last: bool = False):
        self.x = x.to(device=device, dtype=dtype, non_blocking=True).contiguous()
        if channels_last and self.x.dim() == 4:
            self.x = self.x.contiguous(memory_format=torch.channels_last)
        self.y = y.to(device=device, non_blocking=True).contiguous()
        self.device = device
        self.n = self.x.shape[0]
        self.channels_last = channels_last and self.x.dim() == 4

    @classmethod
    def from_loader(cls, loader, device: torch.device, dtype: torch.dtype,
                    channels_last: bool = False) -> "DeviceShard":
        xs, ys = [], []
        for x, y in loader:
            xs.append(x)
            ys.append(y)
        return cls(torch.cat(xs), torch.cat(ys), device, dtype, channels_last)

    def shuffled(self, out_x: Tensor, out_y: Tensor, generator: torch.Generator) -> None:
        """Gather a fresh permutation of the shard into the graph's static
        input buffers (one kernel each)."""
        n = out_x.shape[0]
        perm = torch.randperm(self.n, generator=generator)[:n].to(self.device)
        out_x.copy_(self.x.index_select(0, perm))
        out_y.copy_(self.y.index_select(0, perm))


class TrainGraph:
    """One hipGraph for a full local epoch over ``num_batches`` static batches."""

    def __init__(self, node, shard: DeviceShard, batch_size: int):
        self.node = node
        self.shard = shard
        self.bs = batch_size
        self.num_batches = max(1, shard.n // batch_size)
        n_used = self.num_batches * batch_size
        # zero-init, NOT empty: warmup before the first shuffle must see valid
        # class indices — garbage int64 targets make the NLL gather read out
        # of bounds (an HSA memory fault, found the hard way on the GPU box)
        self.static_x = torch.zeros(
            (n_used, *shard.x.shape[1:]), device=shard.device, dtype=shard.x.dtype
        )
        if shard.channels_last:
            self.static_x = self.static_x.contiguous(memory_format=torch.channels_last)
        self.static_y = torch.zeros((n_used,), device=shard.device, dtype=shard.y.dtype)
        self.loss_sum = torch.zeros((), device=shard.device, dtype=torch.float32)
        # device-resident KL annealing weight for EvidentialLoss capture
        self.kl_weight = torch.zeros((), device=shard.device, dtype=torch.float32)
        self.graph: Optional[torch.cuda.CUDAGraph] = None
        self._params = [p for p in node.model.parameters() if p.requires_grad]
        self._captured_lr: Optional[float] = None
        self._gen = torch.Generator().manual_seed(
            0x5EED ^ (node.node_id * 0x9E3779B9 & 0x7FFFFFFF)
        )

    def _one_step(self, s: int) -> None:
        """One training step inside the capture.

        Gradients flow through ``torch.autograd.grad`` (NOT ``backward``):
        ``backward`` into pre-existing flat-view ``.grad`` buffers records one
        AccumulateGrad ``add_`` kernel PER PARAMETER per step (62 launches per
        ResNet-18 batch — ~300 us of pure kernel-floor cost, measured), while
        ``grad()`` routes each producer kernel's output directly. The SGD
        update is one horizontally-fused ``_foreach_add_`` family over the
        param views (lr is baked into the capture; run_epoch recaptures when
        lr changes — FL configs train at constant lr)."""
        node = self.node
        x = self.static_x[s * self.bs : (s + 1) * self.bs]
        y = self.static_y[s * self.bs : (s + 1) * self.bs]
        from murmura_amd.models.evidential import EvidentialLoss

        out = node.model(x)
        if isinstance(node.criterion, EvidentialLoss):
            loss = node.criterion(out, y, kl_weight=self.kl_weight)
        else:
            loss = torch.nn.functional.cross_entropy(out.float(), y)
        params = self._params
        grads = torch.autograd.grad(loss, params)
        with torch.no_grad():
            torch._foreach_add_(params, grads, alpha=-self._captured_lr)
        self.loss_sum += loss.detach()

    def _capture(self) -> None:
        from murmura_amd.ops.fused_bn import MurmuraBatchNorm2d

        node = self.node
        node.model.train()
        # warmup on a side stream executes real steps — snapshot/restore the
        # flat state so capture-time warmup does not perturb training
        saved = node.store.flat.clone()
        # num_batches_tracked increments are deferred out of the graph (a
        # ~5 us device add per BN per step); local_train bumps them host-side
        # per epoch instead (ops/fused_bn.py bump_num_batches_tracked)
        prev_defer = MurmuraBatchNorm2d.defer_num_batches_tracked
        MurmuraBatchNorm2d.defer_num_batches_tracked = True
        try:
            s = torch.cuda.Stream(device=self.shard.device)
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s):
                for _ in range(2):
                    self.loss_sum.zero_()
                    self._one_step(0)
            torch.cuda.current_stream().wait_stream(s)
            node.store.flat.copy_(saved)
            node.store.zero_grad()
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                self.loss_sum.zero_()
                for step in range(self.num_batches):
                    self._one_step(step)
            self.graph = g
        finally:
            MurmuraBatchNorm2d.defer_num_batches_tracked = prev_defer

    def run_epoch(self, lr: float, round_num: int = 0) -> Tensor:
        """Shuffle + replay one epoch; returns the summed loss (device scalar)."""
        from murmura_amd.models.evidential import EvidentialLoss

        # fill static buffers BEFORE the first capture: warmup executes real
        # kernels and must see genuine inputs/targets
        self.shard.shuffled(self.static_x, self.static_y, self._gen)
        if isinstance(self.node.criterion, EvidentialLoss):
            self.kl_weight.fill_(self.node.criterion.kl_weight_at(round_num))
        if self._captured_lr is not None and lr != self._captured_lr:
            self.graph = None  # lr is baked into the capture; recapture
        if self.graph is None:
            self._captured_lr = float(lr)
            self._capture()
        self.graph.replay()
        return self.loss_sum


class EvalGraph:
    """One hipGraph for a full evaluation pass (CE or evidential)."""

    def __init__(self, node, shard: DeviceShard, batch_size: int, evidential: bool):
        self.node = node
        self.shard = shard
        self.bs = batch_size
        self.evidential = evidential
        self.num_batches = max(1, (shard.n + batch_size - 1) // batch_size)
        # accumulators: [loss_sum, correct] or [vac, ent, str, correct, loss]
        width = 5 if evidential else 2
        self.acc = torch.zeros((width,), device=shard.device, dtype=torch.float32)
        self.graph: Optional[torch.cuda.CUDAGraph] = None

    def _pass(self) -> None:
        node = self.node
        self.acc.zero_()
        for s in range(self.num_batches):
            lo = s * self.bs
            hi = min(self.shard.n, lo + self.bs)
            x = self.shard.x[lo:hi]
            y = self.shard.y[lo:hi]
            out = node.model(x)
            if self.evidential:
                v, e, st, c = ops.evidential_stats(out, y)
                ls, _ = ops.ce_loss_acc(out, y)
                self.acc[0] += v
                self.acc[1] += e
                self.acc[2] += st
                self.acc[3] += c
                self.acc[4] += ls
            else:
                ls, c = ops.ce_loss_acc(out, y)
                self.acc[0] += ls
                self.acc[1] += c

    def _capture(self) -> None:
        self.node.model.eval()
        s = torch.cuda.Stream(device=self.shard.device)
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s), torch.no_grad():
            self._pass()
        torch.cuda.current_stream().wait_stream(s)
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g), torch.no_grad():
            self._pass()
        self.graph = g

    def run(self) -> dict:
        self.node.model.eval()
        if self.graph is None:
            self._capture()
        self.graph.replay()
        n = self.shard.n
        if self.evidential:
            return {
                "vacuity": self.acc[0] / n,
                "entropy": self.acc[1] / n,
                "strength": self.acc[2] / n,
                "accuracy": self.acc[3] / n,
                "loss": self.acc[4] / n,
                "num_samples": n,
            }
        return {"loss": self.acc[0] / n, "accuracy": self.acc[1] / n, "num_samples": n}
