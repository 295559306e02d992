#!/usr/bin/env python3
"""Flagship benchmark: FL rounds/sec, one node per MI355X GPU.

Default config = BASELINE.json config 2: N-node fully-connected FedAvg with a
ResNet-18-sized CNN in bf16 on synthetic 3x32x32 shards (random-init weights —
no network for datasets). Weak scaling: each GPU is one FL node with a fixed
local shard, so per-GPU work is constant as N grows.

One step = one FL round = local epoch(s) of fused-SGD training + neighbor
state exchange over RCCL/xGMI + aggregation + local evaluation (the
reference's default round evaluates every round, core/network.py:80-94 —
nothing is skipped inside the timed region).

Usage:
  python bench.py                          # N=1, quick
  python bench.py --gpus N --steps K --warmup W
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W

Other configs (BASELINE.json configs 3/4):
  --algo krum --attack gaussian            # Krum under 20% Gaussian attackers
  --algo sketchguard --model femnist-xlarge
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--warmup", type=int, default=3)
    p.add_argument("--preset", type=int, default=0, choices=[0, 1, 2, 3, 4, 5],
                   help="BASELINE.json config presets: 1=2-node ring FedAvg tiny "
                        "MLP (CPU plumbing), 2=fully FedAvg resnet18 bf16 "
                        "(default), 3=k-regular(4) Krum + 20%% Gaussian, "
                        "4=Sketchguard ~100M sketch-wire, 5=mobility G^t + "
                        "DMTT/UBAR under topology-liar")
    p.add_argument("--algo", default="fedavg",
                   choices=["fedavg", "krum", "balance", "sketchguard", "ubar",
                            "evidential_trust"])
    p.add_argument("--topology", default=None,
                   help="default: fully (fedavg) / k-regular (krum)")
    p.add_argument("--attack", default="none",
                   choices=["none", "gaussian", "directed", "topology_liar"])
    p.add_argument("--model", default="resnet18",
                   choices=["resnet18", "femnist-baseline", "femnist-xlarge", "mlp",
                            "wide100m", "har"])
    p.add_argument("--sketch-wire", action="store_true",
                   help="sketchguard: exchange 4KB sketches first, full states only with accepted neighbors")
    p.add_argument("--mobility", action="store_true",
                   help="dynamic topology G^t (bounded random walk, seeded)")
    p.add_argument("--dmtt", action="store_true",
                   help="DMTT trust protocol (implies --mobility)")
    p.add_argument("--dtype", default="bf16", choices=["bf16", "fp32"])
    p.add_argument("--shard", type=int, default=2048, help="samples per node")
    p.add_argument("--batch-size", type=int, default=64)
    p.add_argument("--local-epochs", type=int, default=1)
    p.add_argument("--no-eval", action="store_true",
                   help="skip per-round evaluation inside the timed region")
    args = p.parse_args()
    return apply_preset(args)


def apply_preset(args):
    """BASELINE.json's five named configs as one flag each."""
    if args.preset == 1:
        args.algo, args.topology, args.model = "fedavg", "ring", "mlp"
        args.dtype = "fp32"  # CPU plumbing config
    elif args.preset == 2:
        args.algo, args.topology, args.model = "fedavg", "fully", "resnet18"
    elif args.preset == 3:
        args.algo, args.topology, args.model = "krum", "k-regular", "resnet18"
        args.attack = "gaussian"
    elif args.preset == 4:
        args.algo, args.model = "sketchguard", "wide100m"
        args.sketch_wire = True
    elif args.preset == 5:
        args.algo, args.model = "ubar", "har"
        args.attack, args.dmtt = "topology_liar", True
    if args.dmtt:
        args.mobility = True
    return args


def build_config(args, world):
    from murmura_amd.config.schema import Config

    model_cfgs = {
        "resnet18": ("models.resnet18", {"num_classes": 10},
                     {"image_shape": [3, 32, 32], "num_classes": 10}),
        "femnist-baseline": ("models.femnist", {"hidden": 2048},
                             {"image_shape": [1, 28, 28], "num_classes": 62}),
        "femnist-xlarge": ("models.femnist", {"hidden": 8192},
                           {"image_shape": [1, 28, 28], "num_classes": 62}),
        "mlp": ("models.mlp", {"in_features": 32, "hidden": 64, "num_classes": 4},
                {"num_features": 32, "num_classes": 4}),
        # ~100M-param MLP: the Sketchguard bandwidth-bound xGMI showcase
        # (BASELINE.json config 4)
        "wide100m": ("models.widemlp",
                     {"in_features": 4096, "hidden": 12288, "num_classes": 62},
                     {"num_features": 4096, "num_classes": 62}),
        # UCI-HAR-shaped evidential MLP (BASELINE.json config 5 / DMTT)
        "har": ("examples.wearables.uci_har", {"input_dim": 561, "num_classes": 6},
                {"num_features": 561, "num_classes": 6}),
    }
    factory, mparams, dparams = model_cfgs[args.model]
    topo = args.topology or ("k-regular" if args.algo == "krum" else "fully")
    attack_cfg = {"enabled": False}
    if args.attack != "none":
        atk_type = {"gaussian": "gaussian", "directed": "directed_deviation",
                    "topology_liar": "topology_liar"}[args.attack]
        attack_cfg = {
            "enabled": True,
            "type": atk_type,
            "percentage": 0.2,
            "params": ({"model_attack_type": "gaussian", "noise_std": 10.0}
                       if atk_type == "topology_liar" else {"noise_std": 10.0}),
        }
    agg_params = {}
    if args.algo == "krum":
        agg_params = {"num_compromised": max(1, int(0.2 * world))}
    extra = {}
    if args.mobility:
        extra["mobility"] = {"area_size": 100.0, "comm_range": 60.0,
                             "max_speed": 8.0, "seed": 42,
                             "ensure_connected": True}
    if args.dmtt:
        extra["dmtt"] = {}
    return Config(**{
        "experiment": {"name": f"bench-{args.algo}", "seed": 42,
                       "rounds": args.steps + args.warmup, "verbose": False},
        "topology": {"type": topo, "num_nodes": world, "k": 4},
        "aggregation": {"algorithm": args.algo, "params": agg_params},
        "attack": attack_cfg,
        "training": {"local_epochs": args.local_epochs,
                     "batch_size": args.batch_size, "lr": 0.01},
        "data": {"adapter": "synthetic",
                 "params": {"num_samples": args.shard * world,
                            "partition": "iid", **dparams}},
        "model": {"factory": factory, "params": mparams},
        "backend": "rccl",
        "distributed": {"sketch_wire_mode": bool(getattr(args, "sketch_wire", False))},
        "compute": {"dtype": args.dtype, "native_kernels": True},
        **extra,
    })


def main():
    args = parse_args()
    world = int(os.environ.get("WORLD_SIZE", args.gpus))
    rank = int(os.environ.get("RANK", 0))
    launched_distributed = "WORLD_SIZE" in os.environ

    if world > 1 and not launched_distributed:
        print("for N>1 launch via torch.distributed.run", file=sys.stderr)
        sys.exit(2)

    config = build_config(args, world)
    if args.attack != "none" and world == 1 and rank == 0:
        print("warning: with N=1 the single node is selected as compromised "
              "and skips training; attack configs are meaningful for N>=2",
              file=sys.stderr)
    use_cuda = torch.cuda.is_available()

    import torch.distributed as dist

    from murmura_amd.parallel import exchange
    from murmura_amd.parallel.node_process import FLRoundLoop, init_distributed

    if launched_distributed or world > 1:
        device = init_distributed(config, rank, world)
    else:
        # single-process single-node path: no process group needed
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29531")
        dist.init_process_group(
            backend="nccl" if use_cuda else "gloo", rank=0, world_size=1
        )
        device = torch.device("cuda:0" if use_cuda else "cpu")
        if use_cuda:
            torch.cuda.set_device(device)
        exchange.create_host_group_if_needed()

    if config.dmtt is not None:
        from murmura_amd.dmtt.node_process import DMTTRoundLoop

        loop = DMTTRoundLoop(config, rank, world, device)
    else:
        loop = FLRoundLoop(config, rank, world, device)

    pending = [None]

    def one_round(r):
        loop.run_round(r)
        if not args.no_eval:
            # async: this round's training overlapped the PREVIOUS round's
            # eval; resolve it before launching the next (shared accumulator)
            if pending[0] is not None:
                pending[0].resolve()
            pending[0] = loop.evaluate_round_async(r)

    def drain_eval():
        if pending[0] is not None:
            pending[0].resolve()
            pending[0] = None

    def sync():
        if use_cuda:
            torch.cuda.synchronize()
        exchange.barrier(device)
        if use_cuda:
            torch.cuda.synchronize()

    # warmup
    for r in range(args.warmup):
        one_round(r)
    drain_eval()
    sync()
    # timing means should reflect steady state, not capture/comm-init rounds
    loop.timer.totals.clear()
    loop.timer.counts.clear()
    t0 = time.perf_counter()
    for r in range(args.warmup, args.warmup + args.steps):
        one_round(r)
    drain_eval()  # the last round's eval is part of the timed region
    sync()
    elapsed = time.perf_counter() - t0

    # max over ranks
    el = torch.tensor([elapsed], dtype=torch.float64)
    if use_cuda:
        el = el.to(device)
    dist.all_reduce(el, op=dist.ReduceOp.MAX)
    elapsed = float(el.cpu().item())

    ms_per_step = elapsed / args.steps * 1000.0
    rounds_per_sec = args.steps / elapsed

    if os.environ.get("MURMURA_TIMING") == "1" and rank == 0:
        print(f"phase timings (mean ms): {loop.timer.summary()}", file=sys.stderr)

    if rank == 0:
        result = {
            "metric": "FL rounds/sec",
            "value": rounds_per_sec,
            "unit": "rounds/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": args.dtype,
            "data": "synthetic",
            "config": {
                "model": args.model,
                "algorithm": args.algo,
                "topology": config.topology.type,
                "attack": args.attack,
                "global_batch": args.batch_size * world,
                "shard_per_node": args.shard,
                "local_epochs": args.local_epochs,
                "eval_every_round": not args.no_eval,
                "parallelism": f"fl-node-per-gpu x{world} (rccl/xGMI)",
            },
        }
        print(json.dumps(result))
        if getattr(loop, "last_wire_stats", None):
            print(f"sketch-wire stats: {loop.last_wire_stats}", file=sys.stderr)
    exchange.barrier(device)
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
