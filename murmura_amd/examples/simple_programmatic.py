"""Canonical programmatic API usage: ring of 10 nodes, synthetic data, FedAvg
(the reference ships the equivalent example, examples/simple_programmatic.py:43-96).

Run: python -m murmura_amd.examples.simple_programmatic
"""

from __future__ import annotations

import torch
from torch.utils.data import DataLoader

from murmura_amd import FedAvgAggregator, Network, Node, create_topology
from murmura_amd.data.synthetic import load_synthetic_adapter
from murmura_amd.models import SimpleMLP
from murmura_amd.utils.device import get_device
from murmura_amd.utils.seed import set_seed


def main(num_nodes: int = 10, rounds: int = 10) -> None:
    set_seed(42)
    topology = create_topology("ring", num_nodes)
    adapter = load_synthetic_adapter(
        num_nodes, num_samples=2000, num_features=20, num_classes=4, seed=42
    )

    nodes = []
    for i in range(num_nodes):
        device = get_device("auto", i)
        data = adapter.get_client_data(i)
        nodes.append(
            Node(
                node_id=i,
                model=SimpleMLP(20, 32, 4),
                train_loader=DataLoader(data, batch_size=32, shuffle=True),
                test_loader=DataLoader(data, batch_size=64),
                aggregator=FedAvgAggregator(),
                device=device,
                model_factory=lambda: SimpleMLP(20, 32, 4),
            )
        )

    network = Network(nodes, topology)
    history = network.train(rounds=rounds, local_epochs=1, lr=0.05, verbose=True)
    print(f"final mean accuracy: {history['mean_accuracy'][-1]:.4f}")


if __name__ == "__main__":
    main()
