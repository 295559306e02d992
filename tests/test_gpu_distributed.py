"""Multi-rank tests on a single GPU (the only multi-rank GPU config a 1-GPU
lease allows).

- 2 ranks sharing cuda:0 with gloo + host staging (MURMURA_GLOO_CUDA=1):
  validates the device-side compute path (flat stores, HIP kernels, graphs)
  under a real multi-process exchange.
- An nccl co-location probe: RCCL is expected to refuse two ranks on one
  device; the probe records which (skip, not fail) so the audit's claim is
  empirically grounded (docs/NCCL_AUDIT.md §7).

True nccl world>1 runs only on the driver's 8-GPU SCALE pass.
"""

import json
import multiprocessing as mp
import os

import pytest
import torch

pytestmark = pytest.mark.gpu


def _cfg_dict(world, algo="fedavg", topo="ring", port=29721):
    return {
        "experiment": {"name": "gpu2", "seed": 42, "rounds": 2, "verbose": False},
        "topology": {"type": topo, "num_nodes": world},
        "aggregation": {"algorithm": algo},
        "training": {"local_epochs": 1, "batch_size": 16, "lr": 0.05},
        "data": {"adapter": "synthetic",
                 "params": {"num_samples": 80 * world, "num_features": 10,
                            "num_classes": 3}},
        "model": {"factory": "models.mlp",
                  "params": {"in_features": 10, "hidden": 16, "num_classes": 3}},
        "backend": "distributed",
        "distributed": {"comm_backend": "gloo", "master_port": port},
        "compute": {"dtype": "fp32"},
    }


def _gloo_cuda_worker(rank, cfg_json, world, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["MURMURA_GLOO_CUDA"] = "1"
    os.environ["LOCAL_RANK"] = "0"  # both ranks share cuda:0
    os.environ.pop("WORLD_SIZE", None)
    from murmura_amd.config.schema import Config
    from murmura_amd.parallel.node_process import run_node_process

    cfg = Config(**json.loads(cfg_json))
    h = run_node_process(cfg, rank, world)
    if rank == 0:
        q.put(json.dumps({k: v for k, v in h.items() if k != "node_statistics"}))


@pytest.mark.timeout(600)
@pytest.mark.parametrize("algo,topo", [
    ("fedavg", "ring"), ("krum", "fully"), ("ubar", "fully"),
])
def test_two_ranks_one_gpu_gloo_staging(algo, topo):
    port = {"fedavg": 29721, "krum": 29722, "ubar": 29724}[algo]
    cfg = _cfg_dict(2, algo=algo, topo=topo, port=port)
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_gloo_cuda_worker,
                         args=(r, json.dumps(cfg), 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    h = json.loads(q.get(timeout=420))
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    assert len(h["round"]) == 2

    # must match the CPU simulation oracle on the same seeds (ubar's stage-2
    # batch draw differs between backends' loader orders; skip its oracle)
    if algo == "ubar":
        return
    from murmura_amd.cli import _run_simulation as sim
    from murmura_amd.config.schema import Config

    h_sim = sim(Config(**{**cfg, "backend": "simulation",
                          "compute": {"dtype": "fp32", "device": "cpu"}}),
                verbose=False)
    for a, b in zip(h["mean_accuracy"], h_sim["mean_accuracy"]):
        assert abs(a - b) < 2e-2  # fp32 GPU vs CPU tolerance


def _nccl_probe_worker(rank, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["LOCAL_RANK"] = "0"
    import datetime

    import torch
    import torch.distributed as dist

    try:
        dist.init_process_group("nccl", rank=rank, world_size=2,
                                timeout=datetime.timedelta(seconds=60))
        torch.cuda.set_device(0)
        t = torch.ones(4, device="cuda:0")
        dist.all_reduce(t)
        torch.cuda.synchronize()
        ok = bool((t == 2).all().item())
        q.put(("ok", ok))
    except Exception as e:  # RCCL duplicate-GPU refusal expected
        q.put(("err", f"{type(e).__name__}: {e}"))


@pytest.mark.timeout(300)
def test_nccl_colocated_ranks_probe():
    """Empirically record whether RCCL allows 2 ranks on one GPU."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_nccl_probe_worker, args=(r, 29723, q))
             for r in range(2)]
    for p in procs:
        p.start()
    outcomes = []
    try:
        for _ in range(2):
            outcomes.append(q.get(timeout=180))
    except Exception:
        pass
    for p in procs:
        p.join(timeout=30)
        if p.is_alive():
            p.terminate()
            p.join(timeout=10)
    print(f"nccl co-location probe outcomes: {outcomes}")
    if outcomes and all(o[0] == "ok" and o[1] for o in outcomes):
        pass  # RCCL permitted it — even better
    else:
        pytest.skip(f"RCCL refused co-located ranks (expected): {outcomes}")


def _dmtt_worker(rank, cfg_json, world, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["MURMURA_GLOO_CUDA"] = "1"
    os.environ["LOCAL_RANK"] = "0"
    from murmura_amd.config.schema import Config
    from murmura_amd.parallel.node_process import run_node_process

    h = run_node_process(Config(**json.loads(cfg_json)), rank, world)
    if rank == 0:
        q.put(json.dumps({k: v for k, v in h.items() if k != "node_statistics"}))


@pytest.mark.timeout(600)
def test_dmtt_two_ranks_one_gpu():
    """DMTT end-to-end on GPU (2 ranks, gloo staging): claims, trust
    scoring via the captured vmapped forward, TopB selection."""
    cfg = {
        "experiment": {"name": "gpu-dmtt", "seed": 42, "rounds": 3,
                       "verbose": False},
        "topology": {"type": "fully", "num_nodes": 2},
        "aggregation": {"algorithm": "fedavg"},
        "attack": {"enabled": True, "type": "topology_liar", "percentage": 0.5,
                   "params": {"model_attack_type": "gaussian",
                              "noise_std": 10.0}},
        "training": {"local_epochs": 1, "batch_size": 16, "lr": 0.05},
        "data": {"adapter": "synthetic",
                 "params": {"num_samples": 160, "num_features": 20,
                            "num_classes": 4}},
        "model": {"factory": "examples.wearables.uci_har",
                  "params": {"input_dim": 20, "hidden_dims": [16],
                             "num_classes": 4}},
        "backend": "distributed",
        "distributed": {"comm_backend": "gloo", "master_port": 29726},
        "mobility": {"area_size": 100.0, "comm_range": 80.0, "max_speed": 5.0,
                     "seed": 42, "ensure_connected": True},
        "dmtt": {"budget_B": 1},
        "compute": {"dtype": "fp32"},
    }
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_dmtt_worker,
                         args=(r, json.dumps(cfg), 2, 29726, q))
             for r in range(2)]
    for p in procs:
        p.start()
    h = json.loads(q.get(timeout=420))
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    assert len(h["round"]) == 3
