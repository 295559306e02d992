"""Per-node state: model bound to a flat buffer + loaders + aggregator
(reference: murmura/core/node.py:14-252).

Differences from the reference by design:
- The model's float state lives in ONE flat device tensor (core/flat.py);
  ``local_train`` uses the fused SGD kernel (K6) over the flat grad buffer
  instead of building a fresh ``torch.optim.SGD`` per call (node.py:74).
- ``aggregate_with_neighbors`` takes stacked flat states [k, P] and passes an
  EvalContext (scratch flat-bound model) to eval-style aggregators instead of
  injecting train_loader/model_template kwargs (node.py:236-242).
- Evaluation returns device scalars; the host sync happens once per round in
  the Network/metrics layer.
"""

from __future__ import annotations

import os
from typing import Any, Callable, Dict, List, Optional

import torch
from torch import Tensor, nn

from murmura_amd import ops
from murmura_amd.aggregation.base import Aggregator, EvalContext
from murmura_amd.core.flat import FlatParamStore
from murmura_amd.models.evidential import EvidentialLoss
from murmura_amd.utils.metrics import evaluate_evidential, evaluate_model


class Node:
    def __init__(
        self,
        node_id: int,
        model: nn.Module,
        train_loader,
        test_loader,
        aggregator: Aggregator,
        device: torch.device,
        criterion: Optional[nn.Module] = None,
        evidential: bool = False,
        dtype: torch.dtype = torch.float32,
        model_factory: Optional[Callable[[], nn.Module]] = None,
        channels_last: bool = True,
    ) -> None:
        self.node_id = node_id
        self.device = torch.device(device)
        self.dtype = dtype
        # NHWC for conv models on GPU (MIOpen's native layout on CDNA4)
        self.channels_last = (
            channels_last
            and self.device.type == "cuda"
            and any(p.dim() == 4 for p in model.parameters())
        )
        self.store = FlatParamStore(model, self.device, dtype, channels_last=self.channels_last)
        self.model = self.store.model
        self.train_loader = train_loader
        self.test_loader = test_loader
        self.aggregator = aggregator
        self.criterion = criterion if criterion is not None else nn.CrossEntropyLoss()
        self.evidential = evidential
        self.model_factory = model_factory
        self._eval_context: Optional[EvalContext] = None
        # hipGraph round execution (device-resident shard + captured epoch/eval)
        self._shard = None
        self._train_graph = None
        self._eval_graph = None

    # ------------------------------------------------------------ hipGraphs
    def _graphs_enabled(self) -> bool:
        return (
            self.device.type == "cuda"
            and os.environ.get("MURMURA_NO_GRAPHS") != "1"
        )

    def _get_shard(self, split: str = "train"):
        from murmura_amd.core.gpu_round import DeviceShard

        if self._shard is None:
            self._shard = {}
        if split not in self._shard:
            if split == "test" and self.test_loader.dataset is self.train_loader.dataset:
                # reference semantics: test = train data re-served (network.py:289-294)
                self._shard[split] = self._get_shard("train")
            else:
                loader = self.train_loader if split == "train" else self.test_loader
                self._shard[split] = DeviceShard.from_loader(
                    loader, self.device, self.dtype, channels_last=self.channels_last
                )
        return self._shard[split]

    # ------------------------------------------------------------ training
    def local_train(
        self, epochs: int = 1, lr: float = 0.01, round_num: int = 0
    ) -> Dict[str, float]:
        """Local SGD epochs. Skips batches with < 2 samples (BatchNorm needs
        batch > 1 in training mode — reference: node.py:81).

        On GPU with a plain CE criterion the whole epoch runs as ONE hipGraph
        replay over the device-resident shard (core/gpu_round.py); the eager
        path below is the CPU oracle and the evidential/odd-shard fallback.
        """
        if self._graphs_enabled() and isinstance(
            self.criterion, (nn.CrossEntropyLoss, EvidentialLoss)
        ):
            shard = self._get_shard("train")
            bs = getattr(self.train_loader, "batch_size", None) or 32
            if shard.n >= max(2, bs):
                if self._train_graph is None:
                    from murmura_amd.core.gpu_round import TrainGraph

                    self._train_graph = TrainGraph(self, shard, bs)
                total = torch.zeros((), device=self.device)
                for _ in range(max(1, epochs)):
                    total = total + self._train_graph.run_epoch(lr, round_num)
                nb = self._train_graph.num_batches * max(1, epochs)
                # replay the deferred num_batches_tracked increments once per
                # call (the captured graph omits the per-step device adds)
                from murmura_amd.ops.fused_bn import bump_num_batches_tracked

                bump_num_batches_tracked(self.model, nb)
                return {"loss": float(total.item() / nb), "num_batches": nb}
        self.model.train()
        gflat = self.store.ensure_grads()
        pflat = self.store.flat[: self.store.spec.param_numel]
        total_loss = torch.zeros((), device=self.device)
        num_batches = 0
        for _ in range(max(1, epochs)):
            for x, y in self.train_loader:
                if x.shape[0] < 2:
                    continue
                x = x.to(device=self.device, dtype=self.dtype, non_blocking=True)
                if self.channels_last and x.dim() == 4:
                    x = x.contiguous(memory_format=torch.channels_last)
                y = y.to(self.device, non_blocking=True)
                self.store.zero_grad()
                out = self.model(x)
                if isinstance(self.criterion, EvidentialLoss):
                    loss = self.criterion(out, y, round_num=round_num)
                else:
                    loss = self.criterion(out.float(), y)
                loss.backward()
                ops.sgd_step(pflat, gflat, lr)
                total_loss = total_loss + loss.detach()
                num_batches += 1
        mean = (total_loss / max(1, num_batches)).item() if num_batches else 0.0
        return {"loss": float(mean), "num_batches": num_batches}

    # ------------------------------------------------------------ evaluation
    def evaluate(self) -> Dict[str, Tensor]:
        if self._graphs_enabled():
            shard = self._get_shard("test")
            if self._eval_graph is None:
                from murmura_amd.core.gpu_round import EvalGraph

                # evaluation is pure inference: accuracy/loss sums are
                # batch-size independent (BN eval mode uses running stats),
                # so batch as large as the shard allows to fill the 256 CUs
                bs = min(shard.n, 1024)
                self._eval_graph = EvalGraph(self, shard, max(1, bs), self.evidential)
            return self._eval_graph.run()
        if self.evidential:
            return evaluate_evidential(self.model, self.test_loader, self.device, self.dtype)
        return evaluate_model(self.model, self.test_loader, self.device, self.dtype)

    # ------------------------------------------------------------ state access
    def get_state(self) -> Tensor:
        """Pre-round snapshot of the flat state (clone)."""
        return self.store.snapshot()

    def set_state(self, flat: Tensor) -> None:
        self.store.copy_from_flat(flat)

    # ------------------------------------------------------------ aggregation
    def _get_eval_context(self) -> Optional[EvalContext]:
        if self._eval_context is None:
            if self.model_factory is None:
                return None
            scratch = FlatParamStore(
                self.model_factory(), self.device, self.dtype,
                channels_last=self.channels_last,
            )
            self._eval_context = EvalContext(
                scratch, self.train_loader, self.device, self.evidential
            )
        return self._eval_context

    def aggregate_with_neighbors(
        self,
        own_state: Tensor,
        neighbor_states: Tensor,
        neighbor_ids: Optional[List[int]] = None,
        round_num: int = 0,
        **extra: Any,
    ) -> Tensor:
        ctx: Dict[str, Any] = {"neighbor_ids": neighbor_ids, **extra}
        if self.aggregator.requires_eval_context and "eval_context" not in ctx:
            ctx["eval_context"] = self._get_eval_context()
        return self.aggregator.aggregate(
            self.node_id, own_state, neighbor_states, round_num=round_num, **ctx
        )
