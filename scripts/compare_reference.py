"""Head-to-head: reference Murmura vs murmura-amd on IDENTICAL hardware,
config and seeds (the reference publishes no throughput numbers —
BASELINE.md — so this run establishes the baseline ourselves).

Both frameworks run the same experiment through their public programmatic
APIs: N nodes, given topology/aggregator, the same MLP architecture, the same
synthetic shards (561 features / 6 classes, UCI-HAR-shaped), bs 32,
1 local epoch, evaluation every round. Metric: FL rounds/sec over K timed
rounds after W warmup rounds.

Usage:
  python scripts/compare_reference.py [--device cpu|cuda:0] [--nodes 10]
      [--rounds 10] [--algo fedavg|krum|balance]
"""

from __future__ import annotations

import argparse
import os
import sys
import time

import torch
from torch import nn
from torch.utils.data import DataLoader, TensorDataset

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
REFERENCE = "/root/reference"
if not os.path.isdir(REFERENCE):
    REFERENCE = os.path.join(REPO, ".refbench")  # gpurun scratch copy


def make_model():
    return nn.Sequential(
        nn.Linear(561, 256), nn.ReLU(), nn.Linear(256, 128), nn.ReLU(),
        nn.Linear(128, 6),
    )


def make_data(num_nodes: int, per_node: int = 400):
    g = torch.Generator().manual_seed(42)
    n = num_nodes * per_node
    centers = torch.randn(6, 561, generator=g) * 2.0
    y = torch.randint(0, 6, (n,), generator=g)
    x = centers[y] + torch.randn(n, 561, generator=g)
    ds = TensorDataset(x, y)
    parts = [list(range(i * per_node, (i + 1) * per_node)) for i in range(num_nodes)]
    return ds, parts


def run_reference(args, device):
    if not os.path.isdir(REFERENCE):
        print("reference unavailable"); return None
    sys.path.insert(0, REFERENCE)
    from murmura import Network as RefNetwork
    from murmura.aggregation import (
        BALANCEAggregator as RefBalance,
        FedAvgAggregator as RefFedAvg,
        KrumAggregator as RefKrum,
    )
    from murmura.core import Node as RefNode
    from murmura.data import DatasetAdapter as RefAdapter
    from murmura.topology import create_topology as ref_topo
    from murmura.utils import set_seed as ref_seed

    agg = {"fedavg": lambda: RefFedAvg(),
           "krum": lambda: RefKrum(num_compromised=2),
           "balance": lambda: RefBalance()}[args.algo]
    ref_seed(42)
    ds, parts = make_data(args.nodes)
    adapter = RefAdapter(ds, parts)
    topo = ref_topo(args.topology, num_nodes=args.nodes)
    nodes = []
    for i in range(args.nodes):
        sub = adapter.get_client_data(i)
        nodes.append(RefNode(
            node_id=i,
            model=make_model(),
            train_loader=DataLoader(sub, batch_size=32, shuffle=True),
            test_loader=DataLoader(sub, batch_size=32),
            aggregator=agg(),
            device=device,
        ))
    net = RefNetwork(nodes, topo)
    net.train(rounds=args.warmup, local_epochs=1, lr=0.01)
    if device.type == "cuda":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    net.train(rounds=args.rounds, local_epochs=1, lr=0.01)
    if device.type == "cuda":
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    sys.path.remove(REFERENCE)
    for m in [k for k in sys.modules if k == "murmura" or k.startswith("murmura.")]:
        del sys.modules[m]
    return args.rounds / dt


def run_ours(args, device):
    from murmura_amd import Network, Node, create_topology
    from murmura_amd.aggregation import (
        BALANCEAggregator,
        FedAvgAggregator,
        KrumAggregator,
    )
    from murmura_amd.data.adapters import DatasetAdapter
    from murmura_amd.utils.seed import set_seed

    agg = {"fedavg": lambda: FedAvgAggregator(),
           "krum": lambda: KrumAggregator(num_compromised=2),
           "balance": lambda: BALANCEAggregator()}[args.algo]
    set_seed(42)
    ds, parts = make_data(args.nodes)
    adapter = DatasetAdapter(ds, parts)
    topo = create_topology(args.topology, args.nodes)
    nodes = []
    for i in range(args.nodes):
        sub = adapter.get_client_data(i)
        nodes.append(Node(
            node_id=i,
            model=make_model(),
            train_loader=DataLoader(sub, batch_size=32, shuffle=True),
            test_loader=DataLoader(sub, batch_size=32),
            aggregator=agg(),
            device=device,
            model_factory=make_model,
        ))
    net = Network(nodes, topo)
    net.train(rounds=args.warmup, local_epochs=1, lr=0.01)
    if device.type == "cuda":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    net.train(rounds=args.rounds, local_epochs=1, lr=0.01)
    if device.type == "cuda":
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    return args.rounds / dt


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--device", default="cpu")
    ap.add_argument("--nodes", type=int, default=10)
    ap.add_argument("--rounds", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--topology", default="fully")
    ap.add_argument("--algo", default="fedavg",
                    choices=["fedavg", "krum", "balance"])
    ap.add_argument("--skip-reference", action="store_true")
    args = ap.parse_args()
    device = torch.device(args.device)

    ours = run_ours(args, device)
    print(f"murmura-amd  {args.algo:8s} {args.nodes} nodes {args.topology:6s} "
          f"[{args.device}]: {ours:8.3f} rounds/s")
    if not args.skip_reference:
        ref = run_reference(args, device)
        if ref is not None:
            print(f"reference    {args.algo:8s} {args.nodes} nodes {args.topology:6s} "
                  f"[{args.device}]: {ref:8.3f} rounds/s")
            print(f"speedup: {ours / ref:.2f}x")


if __name__ == "__main__":
    main()
