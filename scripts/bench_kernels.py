"""Microbenchmark the HIP aggregation kernels: achieved HBM bandwidth.

Run on the GPU box: python scripts/bench_kernels.py
Each op reports moved-bytes / time vs the ~6.3 TB/s achievable HBM3E ceiling.
"""

import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from murmura_amd import ops


def timeit(fn, n=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n


def report(name, seconds, bytes_moved):
    gbs = bytes_moved / seconds / 1e9
    print(f"{name:46s} {seconds*1e3:9.3f} ms {gbs:10.1f} GB/s")


def main():
    assert ops.native_available()
    dev = torch.device("cuda:0")
    P = 100_000_000  # the ~100M-param Sketchguard config (BASELINE config 4)

    for dtype, esz in [(torch.float32, 4), (torch.bfloat16, 2)]:
        tag = "fp32" if esz == 4 else "bf16"
        m = 8
        x = torch.randn(m, P, device=dev, dtype=dtype)
        w = torch.rand(m, device=dev)

        t = timeit(lambda: ops.weighted_sum(x, w))
        report(f"K1 weighted_sum m=8 P=100M {tag}", t, (m + 1) * P * esz)

        t = timeit(lambda: ops.pairwise_sq_dists(x))
        report(f"K2 pairwise_sq_dists m=8 P=100M {tag}", t, m * P * esz)

        t = timeit(lambda: ops.row_norms(x))
        report(f"K12 row_norms m=8 P=100M {tag}", t, m * P * esz)

        own = x[0].contiguous()
        t = timeit(lambda: ops.l2_dists_to(own, x))
        report(f"K2v l2_dists_to k=8 P=100M {tag}", t, 2 * m * P * esz)

        h, s = ops.make_sketch_tables(P, 1000, 42, dev)
        t = timeit(lambda: ops.count_sketch(x, h, s, 1000))
        # reads: x (m*P*esz) + h (4*P) + s (4*P) per row? h/s re-read per row
        report(f"K4 count_sketch m=8 P=100M S=1000 {tag}", t, m * P * (esz + 8))

        g = torch.randn(P, device=dev, dtype=dtype)
        p = torch.randn(P, device=dev, dtype=dtype)
        t = timeit(lambda: ops.sgd_step(p, g, 1e-9))
        report(f"K6 sgd_step P=100M {tag}", t, 3 * P * esz)

        t = timeit(lambda: ops.gaussian_inject(p, 1.0, 1, 2))
        report(f"K10 gaussian_inject P=100M {tag}", t, 2 * P * esz)

        t = timeit(lambda: ops.scale_inject(p, -5.0))
        report(f"K11 scale_inject P=100M {tag}", t, 2 * P * esz)

        del x, g, p
        torch.cuda.empty_cache()

    # m=11/16 gram (register-pressure cases; m=11 = 10-node reference recipe)
    x = torch.randn(16, 20_000_000, device=dev, dtype=torch.float32)
    t = timeit(lambda: ops.pairwise_sq_dists(x))
    report("K2 pairwise m=16 P=20M fp32", t, 16 * 20_000_000 * 4)
    x11 = x[:11]
    t = timeit(lambda: ops.pairwise_sq_dists(x11))
    report("K2 pairwise m=11 P=20M fp32", t, 11 * 20_000_000 * 4)
    del x, x11
    torch.cuda.empty_cache()

    # eval epilogues
    logits = torch.randn(4096, 62, device=dev, dtype=torch.bfloat16)
    tg = torch.randint(0, 62, (4096,), device=dev)
    t = timeit(lambda: ops.ce_loss_acc(logits, tg))
    report("K7 ce_loss_acc B=4096 C=62 bf16", t, 4096 * 62 * 2)
    t = timeit(lambda: ops.evidential_stats(logits, tg))
    report("K8 evidential_stats B=4096 C=62 bf16", t, 2 * 4096 * 62 * 2)

    # torch-eager comparison for the two hottest ops
    x = torch.randn(8, P // 2, device=dev, dtype=torch.float32)
    w = torch.rand(8, device=dev)
    t_nat = timeit(lambda: ops.weighted_sum(x, w))
    t_ref = timeit(lambda: torch.mv(x.t(), w))
    print(f"weighted_sum native vs torch.mv: {t_nat*1e3:.3f} vs {t_ref*1e3:.3f} ms")
    t_nat = timeit(lambda: ops.pairwise_sq_dists(x))

    def torch_pairwise():
        g = x @ x.t()
        sq = g.diagonal()
        return (sq.unsqueeze(0) + sq.unsqueeze(1) - 2 * g).clamp_min_(0)

    t_ref = timeit(torch_pairwise)
    print(f"pairwise native vs rocBLAS gram: {t_nat*1e3:.3f} vs {t_ref*1e3:.3f} ms")


if __name__ == "__main__":
    main()
