"""Simulation backend: in-process orchestrator with exact synchronous-gossip
round semantics (reference: murmura/core/network.py:16-312).

This backend is the bit-level oracle for the RCCL backend: identical round
semantics (all nodes aggregate against the PRE-ROUND snapshot of every state;
new states apply only after all nodes aggregated — a barrier), identical
seeding, identical history schema. It runs on CPU or on however many GPUs are
visible (nodes pinned round-robin).

MI355X-native details: the pre-round snapshot is ONE stacked [N, P] tensor
per device — aggregation inputs are row views, attack injection is an on-GPU
kernel applied to compromised rows, and per-round metrics stay device-side
until a single host sync at history-append time.
"""

from __future__ import annotations

from typing import Any, Callable, Dict, List, Optional

import torch
from torch import Tensor

from murmura_amd.attacks.base import Attack
from murmura_amd.core.node import Node
from murmura_amd.topology.base import Topology
from murmura_amd.topology.dynamic import MobilityModel
from murmura_amd.topology.generators import create_topology


HISTORY_KEYS = [
    "round",
    "mean_accuracy",
    "std_accuracy",
    "mean_loss",
    "honest_accuracy",
    "compromised_accuracy",
    "mean_vacuity",
    "mean_entropy",
    "mean_strength",
]


def new_history() -> Dict[str, List[float]]:
    """The de-facto metrics API — identical across backends
    (reference: core/network.py:47-58 == distributed/monitor.py:49-59)."""
    return {k: [] for k in HISTORY_KEYS}


class Network:
    def __init__(
        self,
        nodes: List[Node],
        topology: Topology,
        attack: Optional[Attack] = None,
        mobility: Optional[MobilityModel] = None,
    ) -> None:
        if len(nodes) != topology.num_nodes:
            raise ValueError(
                f"{len(nodes)} nodes but topology has {topology.num_nodes}"
            )
        self.nodes = nodes
        self.topology = topology
        self.attack = attack
        self.mobility = mobility
        self.history = new_history()

    # ------------------------------------------------------------ round loop
    def train(
        self,
        rounds: int = 10,
        local_epochs: int = 1,
        lr: float = 0.01,
        eval_every: int = 1,
        verbose: bool = False,
        checkpoint_path: Optional[str] = None,
        checkpoint_every: int = 0,
        start_round: int = 0,
    ) -> Dict[str, List[float]]:
        from murmura_amd.utils import checkpoint as ckpt

        for r in range(start_round, rounds):
            topo = self._topology_at(r)
            self._local_training_step(r, local_epochs, lr)
            self._aggregation_step(r, topo)
            if eval_every and (r % eval_every == 0 or r == rounds - 1):
                self._evaluation_step(r, verbose)
            if checkpoint_path and checkpoint_every and (
                (r + 1) % checkpoint_every == 0 or r == rounds - 1
            ):
                ckpt.save_checkpoint(checkpoint_path, r, self.nodes, self.history)
        return self.history

    def resume_from(self, checkpoint_path: str) -> int:
        """Restore node states + history; returns the next round index."""
        from murmura_amd.utils import checkpoint as ckpt

        return ckpt.restore_network(self, ckpt.load_checkpoint(checkpoint_path))

    def _topology_at(self, round_num: int) -> Topology:
        if self.mobility is not None:
            return self.mobility.topology_at(round_num)
        return self.topology

    def _local_training_step(self, round_num: int, epochs: int, lr: float) -> None:
        """Honest nodes train; compromised nodes skip (frozen models,
        reference: network.py:99-101)."""
        for node in self.nodes:
            if self.attack is not None and self.attack.is_compromised(node.node_id):
                continue
            node.local_train(epochs=epochs, lr=lr, round_num=round_num)

    def _aggregation_step(self, round_num: int, topo: Topology) -> None:
        """Snapshot-once, aggregate-all, apply-after-barrier
        (reference: network.py:105-139)."""
        # one pre-round snapshot per node; on a single shared device this is
        # one stacked [N, P] tensor. The attack alters only the BROADCAST
        # copy — a compromised node's own aggregation uses its clean state
        # (reference: node.py:234 own_state = self.get_state()).
        clean: List[Tensor] = [node.get_state() for node in self.nodes]
        states: List[Tensor] = list(clean)
        if self.attack is not None:
            for i in self.attack.get_compromised_nodes():
                states[i] = self.attack.apply_attack(i, states[i], round_num)

        new_states: List[Optional[Tensor]] = [None] * len(self.nodes)
        for node in self.nodes:
            i = node.node_id
            nbr_ids = topo.neighbors[i]
            if nbr_ids:
                nbr_states = [states[j].to(node.device) for j in nbr_ids]
                stacked = torch.stack(nbr_states, dim=0)
            else:
                stacked = states[i].new_zeros((0, states[i].numel()))
            new_states[i] = node.aggregate_with_neighbors(
                clean[i], stacked, neighbor_ids=nbr_ids, round_num=round_num
            )
        # barrier: apply all aggregated states only after every node aggregated
        for node, ns in zip(self.nodes, new_states):
            node.set_state(ns)

    def _evaluation_step(self, round_num: int, verbose: bool) -> None:
        results = [node.evaluate() for node in self.nodes]
        accs = torch.stack([r["accuracy"].cpu() for r in results]).float()
        losses = torch.stack([r["loss"].cpu() for r in results]).float()
        comp = set(self.attack.get_compromised_nodes()) if self.attack else set()
        honest = [i for i in range(len(self.nodes)) if i not in comp]
        h_acc = accs[honest].mean().item() if honest else 0.0
        c_acc = accs[sorted(comp)].mean().item() if comp else 0.0

        self.history["round"].append(round_num)
        self.history["mean_accuracy"].append(accs.mean().item())
        self.history["std_accuracy"].append(accs.std(unbiased=False).item())
        self.history["mean_loss"].append(losses.mean().item())
        self.history["honest_accuracy"].append(h_acc)
        self.history["compromised_accuracy"].append(c_acc)
        if "vacuity" in results[0]:
            self.history["mean_vacuity"].append(
                torch.stack([r["vacuity"].cpu() for r in results]).mean().item()
            )
            self.history["mean_entropy"].append(
                torch.stack([r["entropy"].cpu() for r in results]).mean().item()
            )
            self.history["mean_strength"].append(
                torch.stack([r["strength"].cpu() for r in results]).mean().item()
            )
        else:
            self.history["mean_vacuity"].append(0.0)
            self.history["mean_entropy"].append(0.0)
            self.history["mean_strength"].append(0.0)
        if verbose:
            print(
                f"[round {round_num}] acc={accs.mean().item():.4f} "
                f"loss={losses.mean().item():.4f} honest={h_acc:.4f}",
                flush=True,
            )

    # ------------------------------------------------------------ statistics
    def get_node_statistics(self) -> Dict[int, Dict[str, Any]]:
        return {n.node_id: n.aggregator.get_statistics() for n in self.nodes}

    # ------------------------------------------------------------ construction
    @classmethod
    def from_config(
        cls,
        config,
        model_factory: Callable[[], torch.nn.Module],
        dataset_adapter,
        aggregator_factory: Callable[[int], Any],
        device: Optional[torch.device] = None,
        criterion_factory: Optional[Callable[[], torch.nn.Module]] = None,
        evidential: bool = False,
    ) -> "Network":
        """Build a Network from a Config (reference: network.py:212-312).

        Per-node DataLoaders clamp the effective batch size to
        min(bs, max(2, n_samples)) with drop_last (reference: :280-287).
        """
        from torch.utils.data import DataLoader

        from murmura_amd.utils.device import get_device
        from murmura_amd.utils.factories import build_attack

        n = config.topology.num_nodes
        topo = create_topology(
            config.topology.type,
            n,
            p=config.topology.p,
            k=config.topology.k,
            seed=config.topology.seed,
        )
        mobility = None
        if config.mobility is not None:
            mobility = MobilityModel(
                n,
                area_size=config.mobility.area_size,
                comm_range=config.mobility.comm_range,
                max_speed=config.mobility.max_speed,
                seed=config.mobility.seed,
                ensure_connected=config.mobility.ensure_connected,
            )
        attack = build_attack(config)
        dtype = torch.bfloat16 if config.compute.dtype == "bf16" else torch.float32
        channels_last = config.compute.channels_last

        def seeded_model(i: int) -> torch.nn.Module:
            # per-node deterministic init: identical weights in the simulation
            # and rccl backends regardless of construction order
            with torch.random.fork_rng(devices=[]):
                torch.manual_seed(config.experiment.seed + i)
                return model_factory()

        nodes: List[Node] = []
        for i in range(n):
            dev = device if device is not None else get_device(config.compute.device, i)
            client_data = dataset_adapter.get_client_data(i)
            n_samples = len(client_data)
            bs = min(config.training.batch_size, max(2, n_samples))
            g = torch.Generator().manual_seed(config.experiment.seed + i)
            train_loader = DataLoader(
                client_data, batch_size=bs, shuffle=True, drop_last=n_samples > bs,
                generator=g,
            )
            # test loader = train data re-served unshuffled (reference:
            # network.py:289-294 — "in practice, separate test set")
            test_loader = DataLoader(client_data, batch_size=bs, shuffle=False)
            criterion = criterion_factory() if criterion_factory else None
            nodes.append(
                Node(
                    node_id=i,
                    model=seeded_model(i),
                    train_loader=train_loader,
                    test_loader=test_loader,
                    aggregator=aggregator_factory(i),
                    device=dev,
                    criterion=criterion,
                    evidential=evidential,
                    dtype=dtype,
                    model_factory=model_factory,
                    channels_last=channels_last,
                )
            )
        return cls(nodes, topo, attack=attack, mobility=mobility)
