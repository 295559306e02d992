"""DMTT 3-condition study, round-2 design.

Round 1's study used BALANCE as the base aggregator; BALANCE alone already
defeats the model attack, so static / mobility / DMTT all scored 0.819 and
the study showed only that the protocol runs. This design gives DMTT
something to do: the base aggregator is plain FedAvg (NO model-side
robustness), attackers run topology-liar wrapping a strong Gaussian model
attack — so the ONLY defense available in condition 3 is DMTT's
trust-scored collaborator selection (excluding attackers from C_i^t via
model vacuity/accuracy scoring + claim verification).

Conditions (reference: documentation/new_murmura_extension/paper.tex:653-713):
  1. static:   fixed fully-connected topology, FedAvg (baseline Murmura)
  2. mobility: dynamic G^t, FedAvg, no trust protocol
  3. dmtt:     same G^t + DMTT trust protocol (TopB collaborator selection)

Run: python experiments/dmtt_study_r2.py [--rounds 30] [--nodes 10]
Writes experiments/results/dmtt_study_r2.json
"""
import argparse
import json
import multiprocessing as mp
import os
import pathlib
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def make_config(condition: str, rounds: int, nodes: int, port: int,
                seed: int = 42) -> dict:
    cfg = {
        "experiment": {"name": f"dmtt-r2-{condition}", "seed": seed,
                       "rounds": rounds, "verbose": False},
        "topology": {"type": "fully", "num_nodes": nodes},
        "aggregation": {"algorithm": "fedavg"},
        "attack": {"enabled": True, "type": "topology_liar", "percentage": 0.3,
                   "params": {"model_attack_type": "directed_deviation",
                              "lambda_param": -5.0}},
        "training": {"local_epochs": 1, "batch_size": 32, "lr": 0.05},
        "data": {"adapter": "synthetic",
                 "params": {"num_samples": 120 * nodes, "num_features": 561,
                            "num_classes": 6, "partition": "dirichlet",
                            "alpha": 0.5}},
        "model": {"factory": "examples.wearables.uci_har",
                  "params": {"input_dim": 561, "hidden_dims": [64],
                             "num_classes": 6}},
        "backend": "distributed",
        "distributed": {"comm_backend": "gloo", "master_port": port},
        "compute": {"dtype": "fp32", "device": "cpu"},
    }
    if condition in ("mobility", "dmtt"):
        cfg["mobility"] = {"area_size": 100.0, "comm_range": 55.0,
                           "max_speed": 8.0, "seed": 42,
                           "ensure_connected": True}
    if condition == "dmtt":
        cfg["dmtt"] = {"budget_B": 4}
    return cfg


def _worker(rank, cfg_json, world, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ.pop("WORLD_SIZE", None)
    from murmura_amd.config.schema import Config
    from murmura_amd.parallel.node_process import run_node_process

    h = run_node_process(Config(**json.loads(cfg_json)), rank, world)
    if rank == 0:
        q.put(json.dumps({k: v for k, v in h.items() if k != "node_statistics"}))


def run_condition(condition, rounds, nodes, port, seed=42):
    cfg = make_config(condition, rounds, nodes, port, seed)
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker,
                         args=(r, json.dumps(cfg), nodes, port, q))
             for r in range(nodes)]
    for p in procs:
        p.start()
    h = json.loads(q.get(timeout=3600))
    for p in procs:
        p.join(timeout=120)
    return h


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rounds", type=int, default=30)
    ap.add_argument("--nodes", type=int, default=10)
    args = ap.parse_args()
    out = {}
    port = 29770
    for cond in ["static", "mobility", "dmtt"]:
        out[cond] = {}
        for seed in (42, 43, 44):
            h = run_condition(cond, args.rounds, args.nodes, port, seed)
            port += 1
            out[cond][str(seed)] = h
            last5 = sum(h["honest_accuracy"][-5:]) / 5
            print(f"{cond:10s} seed={seed} last5 honest = {last5:.4f}", flush=True)
        alls = [sum(v["honest_accuracy"][-5:]) / 5 for v in out[cond].values()]
        mean = sum(alls) / len(alls)
        sd = (sum((a - mean) ** 2 for a in alls) / len(alls)) ** 0.5
        print(f"== {cond:10s} last5 honest mean={mean:.4f} +- {sd:.4f}", flush=True)
    path = pathlib.Path(__file__).parent / "results" / "dmtt_study_r2.json"
    path.write_text(json.dumps(out, indent=1))
    print(f"wrote {path}")


if __name__ == "__main__":
    main()
