"""Probe the slow bs64 ResNet step: per-variant timing (memory format, dtype,
MIOpen find mode) + optional per-kernel top list via torch profiler."""

import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from murmura_amd.models import ResNet18


def timeit(fn, n=10, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1000.0


def step_fn(model, x, y):
    def f():
        for p in model.parameters():
            p.grad = None
        torch.nn.functional.cross_entropy(model(x).float(), y).backward()
    return f


def variant(name, dtype, channels_last, bs, deterministic=False):
    torch.backends.cudnn.deterministic = deterministic
    torch.backends.cudnn.benchmark = not deterministic
    model = ResNet18(num_classes=10).cuda().to(dtype)
    x = torch.randn(bs, 3, 32, 32, device="cuda", dtype=dtype)
    y = torch.randint(0, 10, (bs,), device="cuda")
    if channels_last:
        model = model.to(memory_format=torch.channels_last)
        x = x.to(memory_format=torch.channels_last)
    t = timeit(step_fn(model, x, y))
    print(f"{name:44s} {t:8.2f} ms")
    return t


def main():
    for bs in [64, 256]:
        variant(f"bs{bs} bf16 NCHW benchmark", torch.bfloat16, False, bs)
        variant(f"bs{bs} bf16 NHWC benchmark", torch.bfloat16, True, bs)
        variant(f"bs{bs} fp32 NCHW benchmark", torch.float32, False, bs)
        variant(f"bs{bs} bf16 NCHW deterministic", torch.bfloat16, False, bs, True)

    # per-kernel top list for the slow case
    torch.backends.cudnn.deterministic = False
    torch.backends.cudnn.benchmark = True
    model = ResNet18(10).cuda().to(torch.bfloat16)
    x = torch.randn(64, 3, 32, 32, device="cuda", dtype=torch.bfloat16)
    y = torch.randint(0, 10, (64,), device="cuda")
    f = step_fn(model, x, y)
    for _ in range(5):
        f()
    torch.cuda.synchronize()
    from torch.profiler import ProfilerActivity, profile

    with profile(activities=[ProfilerActivity.CUDA]) as prof:
        for _ in range(5):
            f()
        torch.cuda.synchronize()
    print(prof.key_averages().table(sort_by="self_cuda_time_total", row_limit=18))


if __name__ == "__main__":
    main()
