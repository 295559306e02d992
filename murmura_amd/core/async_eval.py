"""Asynchronous evaluation on a side HIP stream.

Round structure is train -> exchange -> aggregate -> evaluate; evaluation
reads the post-aggregation state that the NEXT round's training immediately
starts mutating, so a naive overlap races. Instead the evaluator owns a
second flat buffer + model: after the round's state is applied, the side
stream snapshots ``store.flat`` into the evaluator's buffer (one D2D copy,
~0.03 ms for the flagship model) and replays its own EvalGraph there while
the default stream proceeds into round r+1's training. The caller resolves
the handle one round later (or at the end), off the critical path.

Used by the distributed round loop and bench; the simulation backend stays
synchronous (it is the semantic oracle).
"""

from __future__ import annotations

from types import SimpleNamespace
from typing import Optional

import torch

from murmura_amd.core.gpu_round import EvalGraph


class EvalHandle:
    def __init__(self, graph: EvalGraph, event: torch.cuda.Event, round_num: int):
        self._graph = graph
        self._event = event
        self.round_num = round_num
        self._result: Optional[dict] = None

    def resolve(self) -> dict:
        """Wait for the side-stream eval and return metric tensors."""
        if self._result is None:
            self._event.synchronize()
            n = self._graph.shard.n
            acc = self._graph.acc
            if self._graph.evidential:
                self._result = {
                    "vacuity": acc[0] / n, "entropy": acc[1] / n,
                    "strength": acc[2] / n, "accuracy": acc[3] / n,
                    "loss": acc[4] / n, "num_samples": n,
                }
            else:
                self._result = {"loss": acc[0] / n, "accuracy": acc[1] / n,
                                "num_samples": n}
        return self._result


class AsyncEvaluator:
    """Owns the snapshot buffer, the side stream, and the captured eval graph."""

    def __init__(self, node):
        from murmura_amd.core.flat import FlatParamStore

        if node.model_factory is None:
            raise ValueError("async eval needs node.model_factory")
        self.node = node
        self.stream = torch.cuda.Stream(device=node.device)
        self.store = FlatParamStore(
            node.model_factory(), node.device, node.dtype,
            channels_last=node.channels_last,
        )
        shard = node._get_shard("test")
        shim = SimpleNamespace(model=self.store.model)
        self.graph = EvalGraph(shim, shard, min(1024, max(1, shard.n)), node.evidential)
        # capture eagerly (inputs are real shard data; weights may be
        # uninitialized, which the eval kernels tolerate) so launch() is a
        # pure copy + replay
        with torch.cuda.stream(self.stream):
            self.store.flat.copy_(node.store.flat)
            self.store.model.eval()
            self.graph._capture()
        torch.cuda.current_stream(node.device).wait_stream(self.stream)

    def launch(self, round_num: int) -> EvalHandle:
        """Snapshot the node's current state and evaluate it on the side
        stream; returns a handle to resolve later."""
        ready = torch.cuda.Event()
        ready.record(torch.cuda.current_stream(self.node.device))
        copied = torch.cuda.Event()
        done = torch.cuda.Event()
        with torch.cuda.stream(self.stream):
            self.stream.wait_event(ready)
            self.store.flat.copy_(self.node.store.flat, non_blocking=True)
            copied.record(self.stream)
            self.graph.graph.replay()
            done.record(self.stream)
        # reverse sync: the default stream must not mutate node.store.flat
        # until the snapshot copy has finished, or the evaluated state could
        # be a half-mutated mix (the eval replay itself may still overlap)
        torch.cuda.current_stream(self.node.device).wait_event(copied)
        return EvalHandle(self.graph, done, round_num)
