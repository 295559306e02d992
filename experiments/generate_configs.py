#!/usr/bin/env python3
"""Generate the full experiment grid (the reference ships ~200 YAMLs produced
the same way — experiments/paper/generate_all_configs.py).

Grid:
- baselines: {uci_har, pamap2, ppg_dalia} x 6 aggregators (paper recipe:
  10 nodes fully-connected, 50 rounds, 2 local epochs, bs 32, dirichlet 0.5)
- attacks:   x {gaussian, directed_deviation} x {10, 20, 30}%
- heterogeneity: dirichlet alpha in {0.1, 0.5, 1.0}
- topologies: {ring, fully, erdos, k-regular}
- dmtt: 3-condition study (static / mobility / mobility+DMTT under
  topology-liar)

``--synthetic`` swaps the wearable adapters for synthetic shards of the same
shape so the grid runs without downloaded datasets (this environment has no
network).

Usage: python experiments/generate_configs.py [--out experiments/configs] [--synthetic]
"""

from __future__ import annotations

import argparse
import itertools
from pathlib import Path

import yaml

DATASETS = {
    "uci_har": {"adapter": "wearables.uci_har", "num_classes": 6,
                "model": "examples.wearables.har_classifier",
                "synth_features": 561, "lr": 0.001},
    "pamap2": {"adapter": "wearables.pamap2", "num_classes": 12,
               "model": "examples.wearables.pamap2_classifier",
               "synth_features": 4000, "lr": 0.001},
    "ppg_dalia": {"adapter": "wearables.ppg_dalia", "num_classes": 7,
                  "model": "examples.wearables.ppg_dalia_classifier",
                  "synth_features": 192, "lr": 0.001},
}
ALGOS = ["fedavg", "krum", "balance", "sketchguard", "ubar", "evidential_trust"]


def base_config(name, ds_key, algo, synthetic, alpha=0.5, topology="fully",
                rounds=50):
    ds = DATASETS[ds_key]
    if synthetic:
        data = {"adapter": "synthetic",
                "params": {"num_samples": 4000, "num_features": ds["synth_features"],
                           "num_classes": ds["num_classes"],
                           "partition": "dirichlet", "alpha": alpha}}
        model = {"factory": ds["model"],
                 "params": {"num_classes": ds["num_classes"]}}
    else:
        data = {"adapter": ds["adapter"],
                "params": {"data_path": f"./data/{ds_key}",
                           "partition_method": "dirichlet", "alpha": alpha}}
        model = {"factory": ds["model"],
                 "params": {"num_classes": ds["num_classes"]}}
    agg_params = {"f": 2} if algo == "krum" else {}
    return {
        "experiment": {"name": name, "seed": 42, "rounds": rounds, "verbose": False},
        "topology": {"type": topology, "num_nodes": 10},
        "aggregation": {"algorithm": algo, "params": agg_params},
        "training": {"local_epochs": 2, "batch_size": 32, "lr": ds["lr"]},
        "data": data,
        "model": model,
        "backend": "simulation",
    }


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--out", default="experiments/configs")
    ap.add_argument("--synthetic", action="store_true")
    ap.add_argument("--rounds", type=int, default=50)
    args = ap.parse_args()
    out = Path(args.out)
    count = 0

    def write(sub, name, cfg):
        nonlocal count
        d = out / sub
        d.mkdir(parents=True, exist_ok=True)
        (d / f"{name}.yaml").write_text(yaml.safe_dump(cfg, sort_keys=False))
        count += 1

    # baselines
    for ds, algo in itertools.product(DATASETS, ALGOS):
        cfg = base_config(f"baseline_{ds}_{algo}", ds, algo, args.synthetic,
                          alpha=0.1, rounds=args.rounds)
        write("baseline", f"{ds}_{algo}", cfg)

    # attacks
    for ds, algo, atk, pct in itertools.product(
        DATASETS, ALGOS, ["gaussian", "directed_deviation"], [10, 20, 30]
    ):
        cfg = base_config(f"attacks_{ds}_{algo}_{atk}_{pct}pct", ds, algo,
                          args.synthetic, alpha=0.1, rounds=args.rounds)
        cfg["attack"] = {"enabled": True, "type": atk, "percentage": pct / 100.0,
                         "params": {"noise_std": 10.0} if atk == "gaussian"
                         else {"deviation_factor": -5.0}}
        write(f"attacks/{ds}", f"{algo}_{atk}_{pct}pct", cfg)

    # heterogeneity
    for ds, algo, alpha in itertools.product(DATASETS, ALGOS, [0.1, 0.5, 1.0]):
        cfg = base_config(f"hetero_{ds}_{algo}_a{alpha}", ds, algo,
                          args.synthetic, alpha=alpha, rounds=args.rounds)
        write(f"heterogeneity/{ds}", f"{algo}_alpha{alpha}", cfg)

    # topologies
    for ds, algo, topo in itertools.product(
        ["uci_har"], ALGOS, ["ring", "fully", "erdos", "k-regular"]
    ):
        cfg = base_config(f"topo_{ds}_{algo}_{topo}", ds, algo, args.synthetic,
                          topology=topo, rounds=args.rounds)
        write("topologies", f"{algo}_{topo}", cfg)

    # dmtt 3-condition study (balance aggregation: a defended baseline so the
    # trust protocol's collaborator selection has a measurable effect)
    for cond in ["static", "mobility", "dmtt"]:
        cfg = base_config(f"dmtt_{cond}", "uci_har", "balance", args.synthetic,
                          rounds=min(args.rounds, 30))
        cfg["topology"]["type"] = "ring"
        cfg["backend"] = "distributed"
        cfg["attack"] = {"enabled": True, "type": "topology_liar",
                         "percentage": 0.2,
                         "params": {"model_attack_type": "gaussian",
                                    "noise_std": 5.0}}
        if cond in ("mobility", "dmtt"):
            cfg["mobility"] = {"area_size": 100.0, "comm_range": 40.0,
                               "max_speed": 5.0, "seed": 42}
        if cond == "dmtt":
            cfg["dmtt"] = {"budget_B": 5}
        write("dmtt", f"{cond}", cfg)

    print(f"wrote {count} configs under {out}")


if __name__ == "__main__":
    main()
