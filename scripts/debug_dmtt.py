"""Run one DMTT-study condition with optional debug printing.

Usage: python scripts/debug_dmtt.py [condition] [rounds] [port]
"""
import json
import multiprocessing as mp
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from experiments.dmtt_study_r2 import make_config  # noqa: E402


def worker(rank, cfg_json, world, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    from murmura_amd.config.schema import Config
    from murmura_amd.parallel.node_process import run_node_process

    h = run_node_process(Config(**json.loads(cfg_json)), rank, world)
    if rank == 0 and q is not None:
        q.put(json.dumps({k: v for k, v in h.items() if k != "node_statistics"}))


def main():
    cond = sys.argv[1] if len(sys.argv) > 1 else "dmtt"
    rounds = int(sys.argv[2]) if len(sys.argv) > 2 else 8
    port = int(sys.argv[3]) if len(sys.argv) > 3 else 29790
    cfg = make_config(cond, rounds, 10, port)
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=worker, args=(r, json.dumps(cfg), 10, port, q))
             for r in range(10)]
    for p in procs:
        p.start()
    h = json.loads(q.get(timeout=900))
    for p in procs:
        p.join(timeout=120)
    print("honest:", [round(a, 3) for a in h["honest_accuracy"]])
    print("loss:  ", [round(a, 3) for a in h["mean_loss"]])


if __name__ == "__main__":
    main()
