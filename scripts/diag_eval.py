"""Diagnose the round-2 bench regression: time train vs eval paths."""
import os
import sys
import time

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from bench import apply_preset, build_config  # noqa: E402


class A:
    preset = 0
    algo = "fedavg"; topology = None; attack = "none"; model = "resnet18"
    sketch_wire = False; mobility = False; dmtt = False
    dtype = "bf16"; shard = 2048; batch_size = 64; local_epochs = 1
    no_eval = False; gpus = 1; steps = 5; warmup = 2


def main():
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29541")
    dist.init_process_group("nccl" if torch.cuda.is_available() else "gloo",
                            rank=0, world_size=1)
    cfg = build_config(A(), 1)
    from murmura_amd.parallel.node_process import FLRoundLoop

    dev = torch.device("cuda:0")
    torch.cuda.set_device(dev)
    loop = FLRoundLoop(cfg, 0, 1, dev)

    def t(label, fn, n=3):
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(n):
            r = fn()
        torch.cuda.synchronize()
        print(f"{label}: {(time.perf_counter()-t0)/n*1000:.2f} ms", flush=True)
        return r

    # warm everything once
    loop.run_round(0)
    t("run_round (train+agg only)", lambda: loop.run_round(1), 3)

    # sync eval graph path
    t("node.evaluate() first (capture)", lambda: loop.node.evaluate(), 1)
    t("node.evaluate() steady", lambda: loop.node.evaluate(), 3)

    # async evaluator
    from murmura_amd.core.async_eval import AsyncEvaluator

    t0 = time.perf_counter()
    try:
        ae = AsyncEvaluator(loop.node)
        torch.cuda.synchronize()
        print(f"AsyncEvaluator ctor: {(time.perf_counter()-t0)*1000:.2f} ms", flush=True)
        h = ae.launch(0)
        t("async resolve steady", lambda: ae.launch(1).resolve(), 3)
        _ = h.resolve()
    except Exception as e:
        print(f"AsyncEvaluator FAILED: {type(e).__name__}: {e}", flush=True)

    # full pipelined round as bench does it
    pending = [None]

    def one(r):
        loop.run_round(r)
        if pending[0] is not None:
            pending[0].resolve()
        pending[0] = loop.evaluate_round_async(r)

    one(2)
    t("pipelined round steady", lambda: one(3), 5)
    print("async eval type:", type(getattr(loop, "_async_eval", None)), flush=True)
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
