"""Aggregate-phase timing at m=8 on one GPU via the simulation backend
(8 logical nodes, k=7 neighbors each — the N=1 distributed bench has no
neighbors, so this is where UBAR/EvidentialTrust candidate scoring shows).

VERDICT round-1 item 3 'Done' check: aggregate-phase time for ubar /
evidential_trust comparable to krum's.
"""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from murmura_amd.config.schema import Config
from murmura_amd.core.network import Network
from murmura_amd.utils import factories
from murmura_amd.utils.seed import set_seed


def build(algo):
    cfg = Config(**{
        "experiment": {"name": f"agg-{algo}", "seed": 42, "rounds": 10,
                       "verbose": False},
        "topology": {"type": "fully", "num_nodes": 8},
        "aggregation": {"algorithm": algo},
        "training": {"local_epochs": 1, "batch_size": 32, "lr": 0.01},
        "data": {"adapter": "synthetic",
                 "params": {"num_samples": 1600, "num_features": 561,
                            "num_classes": 6}},
        "model": {"factory": "examples.wearables.uci_har",
                  "params": {"input_dim": 561, "num_classes": 6}},
        "compute": {"dtype": "fp32"},
    })
    set_seed(42)
    mf = factories.build_model_factory(cfg)
    dev = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
    return Network.from_config(
        cfg, mf, factories.build_dataset_adapter(cfg),
        factories.build_aggregator_factory(cfg, mf), device=dev,
        criterion_factory=factories.build_criterion_factory(cfg),
        evidential=factories.is_evidential(cfg),
    )


def main():
    for algo in ["fedavg", "krum", "balance", "sketchguard", "ubar",
                 "evidential_trust"]:
        net = build(algo)
        topo = net.topology
        # warmup (captures, vmap compile, MIOpen find)
        net._local_training_step(0, 1, 0.01)
        net._aggregation_step(0, topo)
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        n = 5
        for r in range(1, 1 + n):
            net._aggregation_step(r, topo)
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / n / 8  # per node per round
        print(f"{algo:18s} aggregate (m=8, per node): {dt*1e3:8.3f} ms")


if __name__ == "__main__":
    main()
