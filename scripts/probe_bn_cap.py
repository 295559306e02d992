"""Sweep BN reduction kernel time by shape (cap set via MURMURA_BN_GRID_CAP)."""
import os, sys, time
import torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from murmura_amd import ops

def timeit(fn, n=50, warmup=10):
    for _ in range(warmup): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e6  # us

ext = ops._load_ext()
cap = os.environ.get("MURMURA_BN_GRID_CAP", "256")
for c, hw in [(64, 32), (128, 16), (256, 8), (512, 4)]:
    x = torch.randn(64, c, hw, hw, device="cuda", dtype=torch.bfloat16).contiguous(
        memory_format=torch.channels_last)
    w = torch.ones(c, device="cuda", dtype=torch.bfloat16)
    b = torch.zeros_like(w)
    rm = torch.zeros_like(w); rv = torch.ones_like(w)
    t_f = timeit(lambda: ext.bn_fwd_train(x, w, b, rm, rv, 0.1, 1e-5, True))
    y, mean, invstd = ext.bn_fwd_train(x, w, b, rm, rv, 0.1, 1e-5, True)
    dy = torch.randn_like(x).contiguous(memory_format=torch.channels_last)
    t_b = timeit(lambda: ext.bn_bwd(x, dy, w, mean, invstd, y, True))
    mb = x.numel() * 2 / 1e6
    print(f"cap={cap:>4} C={c:4d} HW={hw:2d} ({mb:5.1f}MB): fwd {t_f:7.1f}us  bwd {t_b:7.1f}us")
