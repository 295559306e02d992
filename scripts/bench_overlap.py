"""Measure the exchange-overlap and sketch-wire claims.

Two ranks on one GPU via gloo + host staging (the only multi-rank config a
1-GPU lease allows): the CHUNKED path's Gram-while-in-flight overlap cannot
show its xGMI benefit here (gloo staging serializes the wire through host
memory — stated on every row), but this measures (a) that chunking costs
nothing when it cannot help, (b) the sketch-wire mode's measured byte
savings and round rate, (c) allreduce vs P2P full-state exchange.

Run: python scripts/bench_overlap.py  (spawns 2 worker processes)
"""
import json
import multiprocessing as mp
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

MODEL = {"factory": "models.widemlp",
         "params": {"in_features": 4096, "hidden": 4096, "num_classes": 62}}
# ~34M params fp32 -> 136 MB per state on the wire


def _cfg(algo, overlap, sketch_wire, port, world=2, attack=False):
    d = {
        "experiment": {"name": "ovl", "seed": 42, "rounds": 1, "verbose": False},
        "topology": {"type": "fully", "num_nodes": world},
        "aggregation": {"algorithm": algo},
        "training": {"local_epochs": 1, "batch_size": 32, "lr": 0.01},
        "data": {"adapter": "synthetic",
                 "params": {"num_samples": 64 * world, "num_features": 4096,
                            "num_classes": 62}},
        "model": MODEL,
        "backend": "distributed",
        "distributed": {"comm_backend": "gloo", "master_port": port,
                        "overlap_exchange": overlap,
                        "sketch_wire_mode": sketch_wire},
        "compute": {"dtype": "fp32"},
    }
    if attack:
        d["attack"] = {"enabled": True, "type": "gaussian", "percentage": 0.3,
                       "params": {"noise_std": 50.0}}
    return d


def _worker(rank, cfg_json, port, q, world):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["MURMURA_GLOO_CUDA"] = "1"
    os.environ["LOCAL_RANK"] = "0"
    import torch

    from murmura_amd.config.schema import Config
    from murmura_amd.parallel.node_process import FLRoundLoop, init_distributed

    cfg = Config(**json.loads(cfg_json))
    device = init_distributed(cfg, rank, world)
    loop = FLRoundLoop(cfg, rank, world, device)
    loop.run_round(0)  # warmup: captures, comm setup
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    n = 5
    t0 = time.perf_counter()
    for r in range(1, 1 + n):
        loop.run_round(r)
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / n
    if rank == 0:
        q.put({
            "ms_per_round": dt * 1e3,
            "exchange_ms": loop.timer.summary().get("exchange"),
            "wire_stats": getattr(loop, "last_wire_stats", None),
        })
    import torch.distributed as dist

    dist.barrier()
    dist.destroy_process_group()


def run(algo, overlap, sketch_wire, port, world=2, attack=False):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    cfg = _cfg(algo, overlap, sketch_wire, port, world, attack)
    procs = [ctx.Process(target=_worker, args=(r, json.dumps(cfg), port, q, world))
             for r in range(world)]
    for p in procs:
        p.start()
    res = q.get(timeout=300)
    for p in procs:
        p.join(timeout=60)
    return res


def main():
    port = 29751
    rows = []
    for algo, overlap, wire, world, attack, label in [
        ("krum", False, False, 2, False, "krum P2P unchunked w2"),
        ("krum", True, False, 2, False, "krum chunked+gram-overlap w2"),
        ("fedavg", False, False, 2, False, "fedavg allreduce fast path w2"),
        ("sketchguard", False, False, 4, True, "sketchguard full exchange w4+atk"),
        ("sketchguard", False, True, 4, True, "sketchguard sketch-wire w4+atk"),
    ]:
        r = run(algo, overlap, wire, port, world, attack)
        port += 1
        rows.append((label, r))
        print(f"{label:32s} {r['ms_per_round']:8.1f} ms/round "
              f"exchange={r['exchange_ms']}  wire={r['wire_stats']}", flush=True)
    print(json.dumps({k: v for k, v in rows}, default=str))


if __name__ == "__main__":
    main()
