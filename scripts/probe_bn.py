"""A/B: ResNet-18 step with fused MurmuraBatchNorm2d vs nn.BatchNorm2d,
plus per-kernel profile of the fused-BN step."""

import os
import sys
import time

import torch
from torch import nn

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from murmura_amd.models import ResNet18
from murmura_amd.ops.fused_bn import MurmuraBatchNorm2d


def timeit(fn, n=20, warmup=8):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1000.0


def swap_bn(model, cls):
    for name, mod in model.named_children():
        if isinstance(mod, nn.BatchNorm2d):
            new = cls(mod.num_features).to(mod.weight.device, mod.weight.dtype)
            setattr(model, name, new)
        else:
            swap_bn(mod, cls)
    return model


def bench_model(model, tag):
    model = model.cuda().to(torch.bfloat16).to(memory_format=torch.channels_last)
    x = torch.randn(64, 3, 32, 32, device="cuda", dtype=torch.bfloat16).contiguous(
        memory_format=torch.channels_last
    )
    y = torch.randint(0, 10, (64,), device="cuda")
    model.train()

    def step():
        for p in model.parameters():
            p.grad = None
        torch.nn.functional.cross_entropy(model(x).float(), y).backward()

    t = timeit(step)
    print(f"{tag:34s} {t:7.3f} ms/step")
    return step


def main():
    torch.backends.cudnn.benchmark = True
    # sanity: how many fused BN modules and do they take the fused path?
    m = ResNet18(10)
    kinds = {}
    for mod in m.modules():
        kinds[type(mod).__name__] = kinds.get(type(mod).__name__, 0) + 1
    print("module census:", {k: v for k, v in kinds.items() if "atch" in k})

    step_fused = bench_model(ResNet18(10), "fused MurmuraBatchNorm2d")
    bench_model(swap_bn(ResNet18(10), nn.BatchNorm2d), "plain nn.BatchNorm2d")

    # isolated BN op A/B at the hot shape
    for c, hw in [(64, 32), (128, 16), (256, 8), (512, 4)]:
        x = torch.randn(64, c, hw, hw, device="cuda", dtype=torch.bfloat16).contiguous(
            memory_format=torch.channels_last
        ).requires_grad_(True)
        fused = MurmuraBatchNorm2d(c).cuda().to(torch.bfloat16)
        plain = nn.BatchNorm2d(c).cuda().to(torch.bfloat16)
        g = torch.randn_like(x)

        def run(mod):
            def f():
                x.grad = None
                mod(x).backward(g)
            return f

        tf = timeit(run(fused))
        tp = timeit(run(plain))
        print(f"BN C={c:4d} HW={hw:3d}: fused {tf:7.3f} ms  torch {tp:7.3f} ms  ({tp/tf:.2f}x)")

    from torch.profiler import ProfilerActivity, profile

    with profile(activities=[ProfilerActivity.CUDA]) as prof:
        for _ in range(5):
            step_fused()
        torch.cuda.synchronize()
    print(prof.key_averages().table(sort_by="self_cuda_time_total", row_limit=15))


if __name__ == "__main__":
    main()
