"""Phase-level timing of one FL round on GPU — where do the milliseconds go?

Run on the GPU box:  python scripts/profile_round.py
"""

import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def timeit(fn, n=5, warmup=2):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1000.0  # ms


def build_loop(dtype="bf16", deterministic=True, graphs=True):
    os.environ.pop("MURMURA_NO_GRAPHS", None)
    if not graphs:
        os.environ["MURMURA_NO_GRAPHS"] = "1"
    import torch.distributed as dist

    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29581")
        dist.init_process_group(backend="nccl", rank=0, world_size=1)
    from bench import build_config
    from murmura_amd.parallel.node_process import FLRoundLoop

    class A:
        pass

    args = A()
    args.algo = "fedavg"
    args.topology = None
    args.attack = "none"
    args.model = "resnet18"
    args.dtype = dtype
    args.shard = 2048
    args.batch_size = 64
    args.local_epochs = 1
    args.steps, args.warmup = 1, 1
    cfg = build_config(args, 1)
    loop = FLRoundLoop(cfg, 0, 1, torch.device("cuda:0"))
    torch.backends.cudnn.deterministic = deterministic
    torch.backends.cudnn.benchmark = not deterministic
    return loop


def main():
    loop = build_loop()
    node = loop.node

    # warm everything (captures graphs)
    loop.run_round(0)
    loop.evaluate_round(0)

    t_train = timeit(lambda: node.local_train(1, 0.01, 1))
    t_eval = timeit(lambda: node.evaluate())
    t_round = timeit(lambda: loop.run_round(1))
    t_evalr = timeit(lambda: loop.evaluate_round(1))
    print(f"graph path: local_train={t_train:.1f}ms evaluate={t_eval:.1f}ms "
          f"run_round={t_round:.1f}ms evaluate_round={t_evalr:.1f}ms")

    # inner pieces
    tg = node._train_graph
    t_shuffle = timeit(lambda: tg.shard.shuffled(tg.static_x, tg.static_y, tg._gen))
    t_replay = timeit(lambda: tg.graph.replay())
    eg = node._eval_graph
    t_ereplay = timeit(lambda: eg.graph.replay())
    print(f"pieces: shuffle={t_shuffle:.1f}ms train_replay={t_replay:.1f}ms "
          f"eval_replay={t_ereplay:.1f}ms")

    # benchmark-mode (non-deterministic MIOpen) comparison
    torch.backends.cudnn.deterministic = False
    torch.backends.cudnn.benchmark = True
    t_replay_b = timeit(lambda: tg.graph.replay(), n=10)
    print(f"replay after benchmark-mode flags (same graph): {t_replay_b:.1f}ms")

    # eager fwd/bwd single batch for reference
    x = torch.randn(64, 3, 32, 32, device="cuda", dtype=node.dtype)
    y = torch.randint(0, 10, (64,), device="cuda")

    def one_batch():
        node.store.zero_grad()
        loss = torch.nn.functional.cross_entropy(node.model(x).float(), y)
        loss.backward()

    node.model.train()
    t_batch = timeit(one_batch, n=10, warmup=5)
    print(f"eager fwd+bwd one batch bs64: {t_batch:.2f}ms -> epoch(32)~{32*t_batch:.0f}ms")

    # bigger batch utilization probe
    for bs in [256, 1024]:
        xb = torch.randn(bs, 3, 32, 32, device="cuda", dtype=node.dtype)
        yb = torch.randint(0, 10, (bs,), device="cuda")

        def fb():
            node.store.zero_grad()
            torch.nn.functional.cross_entropy(node.model(xb).float(), yb).backward()

        print(f"eager fwd+bwd bs{bs}: {timeit(fb, n=5, warmup=3):.2f}ms")


if __name__ == "__main__":
    main()
