"""8-rank gloo CPU rehearsal of the full bench path.

The driver's SCALE pass runs `bench.py --gpus 8` on a box this code has never
touched; this test runs the SAME round-loop code at world_size 8 over gloo —
allreduce fast path, chunked Krum exchange+Gram overlap, sketch-wire
Sketchguard, DMTT under topology-liar, and the straggler ready-bit — so the
only untested delta on the real box is the nccl transport itself (audited in
docs/NCCL_AUDIT.md).
"""

import json
import multiprocessing as mp
import os

import pytest

WORLD = 8


def _worker(rank, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ.pop("WORLD_SIZE", None)

    from murmura_amd.config.schema import Config
    from murmura_amd.parallel.node_process import run_node_process

    def cfg(algo, topo="fully", attack=None, extra=None, model_extra=None):
        d = {
            "experiment": {"name": "g8", "seed": 42, "rounds": 2, "verbose": False},
            "topology": {"type": topo, "num_nodes": WORLD, "k": 4},
            "aggregation": {"algorithm": algo},
            "training": {"local_epochs": 1, "batch_size": 16, "lr": 0.05},
            "data": {"adapter": "synthetic",
                     "params": {"num_samples": 40 * WORLD, "num_features": 10,
                                "num_classes": 3}},
            "model": {"factory": "models.mlp",
                      "params": {"in_features": 10, "hidden": 8, "num_classes": 3}},
            "backend": "distributed",
            "distributed": {"comm_backend": "gloo", "master_port": port},
        }
        if attack:
            d["attack"] = attack
        if extra:
            d.update(extra)
        if model_extra:
            d["model"] = model_extra
        return Config(**d)

    results = {}

    # 1. fedavg + fully-connected: the allreduce fast path at world 8
    h = run_node_process(cfg("fedavg"), rank, WORLD, destroy_group=False)
    if rank == 0:
        results["fedavg_fully"] = h

    # 2. krum on k-regular(4) with 20% gaussian attackers: chunked
    #    exchange + Gram overlap path (overlap_exchange defaults True)
    h = run_node_process(
        cfg("krum", topo="k-regular",
            attack={"enabled": True, "type": "gaussian", "percentage": 0.2,
                    "params": {"noise_std": 20.0}}),
        rank, WORLD, destroy_group=False,
    )
    if rank == 0:
        results["krum_kreg_attack"] = h

    # 3. sketchguard in sketch-wire mode: sketch all-gather + want-mask
    #    symmetrization + partial full-state exchange
    h = run_node_process(
        cfg("sketchguard",
            extra={"distributed": {"comm_backend": "gloo", "master_port": port,
                                   "sketch_wire_mode": True}}),
        rank, WORLD, destroy_group=False,
    )
    if rank == 0:
        results["sketchguard_wire"] = h

    # 4. straggler ready-bit: generous budget so nobody is dropped, but the
    #    per-round host-group all_gather of ready bits executes at world 8
    h = run_node_process(
        cfg("balance",
            topo="ring",
            extra={"distributed": {"comm_backend": "gloo", "master_port": port,
                                   "round_duration_s": 3600.0}}),
        rank, WORLD, destroy_group=False,
    )
    if rank == 0:
        results["balance_ready_bit"] = h

    # 5. DMTT + mobility + topology-liar + evidential model: want-mask
    #    symmetrization, claims all-gather, trust scoring at world 8
    h = run_node_process(
        cfg("fedavg",
            attack={"enabled": True, "type": "topology_liar", "percentage": 0.25,
                    "params": {"model_attack_type": "gaussian", "noise_std": 10.0}},
            extra={
                "mobility": {"area_size": 100.0, "comm_range": 60.0,
                             "max_speed": 8.0, "seed": 42,
                             "ensure_connected": True},
                "dmtt": {},
            },
            model_extra={"factory": "examples.wearables.uci_har",
                         "params": {"input_dim": 10, "hidden_dims": [16],
                                    "num_classes": 3}}),
        rank, WORLD, destroy_group=True,
    )
    if rank == 0:
        results["dmtt"] = h
        q.put(json.dumps({k: {kk: vv for kk, vv in v.items()
                              if kk != "node_statistics"}
                          for k, v in results.items()}))


@pytest.mark.timeout(600)
def test_world8_full_bench_path():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, 29671, q)) for r in range(WORLD)]
    for p in procs:
        p.start()
    payload = json.loads(q.get(timeout=540))
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0, f"rank process exited {p.exitcode}"

    assert set(payload) == {
        "fedavg_fully", "krum_kreg_attack", "sketchguard_wire",
        "balance_ready_bit", "dmtt",
    }
    for name, h in payload.items():
        assert len(h["round"]) == 2, name
        assert all(0.0 <= a <= 1.0 for a in h["mean_accuracy"]), name

    # fedavg fully-connected at world 8 must match the simulation oracle
    from murmura_amd.cli import _run_simulation as sim
    from murmura_amd.config.schema import Config

    sim_cfg = Config(**{
        "experiment": {"name": "g8", "seed": 42, "rounds": 2, "verbose": False},
        "topology": {"type": "fully", "num_nodes": WORLD},
        "aggregation": {"algorithm": "fedavg"},
        "training": {"local_epochs": 1, "batch_size": 16, "lr": 0.05},
        "data": {"adapter": "synthetic",
                 "params": {"num_samples": 40 * WORLD, "num_features": 10,
                            "num_classes": 3}},
        "model": {"factory": "models.mlp",
                  "params": {"in_features": 10, "hidden": 8, "num_classes": 3}},
        "backend": "simulation",
    })
    h_sim = sim(sim_cfg, verbose=False)
    for a, b in zip(payload["fedavg_fully"]["mean_accuracy"],
                    h_sim["mean_accuracy"]):
        assert abs(a - b) < 1e-4
