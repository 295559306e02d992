"""Single-machine launcher for the RCCL backend: spawns one process per node
(reference: murmura/distributed/runner.py:114-213, minus the ZMQ monitor — the
rank-0 process collects metrics through the collective gather path instead of
a separate monitor process)."""

from __future__ import annotations

import os
from typing import Dict, List, Optional

import torch
import torch.multiprocessing as mp

from murmura_amd.config.schema import Config


def _worker(rank: int, config_json: str, world_size: int, queue) -> None:
    from murmura_amd.config.schema import Config as _C
    from murmura_amd.parallel.node_process import run_node_process

    config = _C.model_validate_json(config_json)
    history = run_node_process(config, rank, world_size)
    if rank == 0:
        queue.put(history)


class DistributedRunner:
    def __init__(self, config: Config):
        self.config = config

    def run(self) -> Dict[str, List[float]]:
        world = self.config.topology.num_nodes
        if torch.cuda.is_available() and world > torch.cuda.device_count():
            raise RuntimeError(
                f"{world} nodes but only {torch.cuda.device_count()} GPUs; "
                "one node maps to one GPU in the rccl backend"
            )
        os.environ["MASTER_ADDR"] = self.config.distributed.master_addr
        # per-launch port offset: two concurrent runners on one host must not
        # share a rendezvous port (cross-talk partially formed both worlds)
        port = self.config.distributed.master_port + (os.getpid() % 997)
        os.environ["MASTER_PORT"] = str(port)
        # avoid CPU thread oversubscription: N worker processes on one host
        if not torch.cuda.is_available():
            os.environ.setdefault("OMP_NUM_THREADS", "2")
        ctx = mp.get_context("spawn")
        queue = ctx.SimpleQueue()
        cfg_json = self.config.model_dump_json()
        procs = mp.spawn(
            _worker,
            args=(cfg_json, world, queue),
            nprocs=world,
            join=False,
        )
        history: Optional[Dict[str, List[float]]] = None
        # rank 0 puts history before the final barrier; read it, then join
        history = queue.get()
        procs.join()
        return history
