"""Measure the per-kernel wall floor: tiny kernels back-to-back, eager vs graph."""
import sys, os, time
sys.path.insert(0, '/root/repo')
import torch

dev = torch.device("cuda")
x = torch.zeros(512, device=dev)
y = torch.zeros(512, device=dev)

def run_eager(n):
    for _ in range(n):
        torch.add(x, 1.0, out=y)

# graph with 1000 tiny adds
g = torch.cuda.CUDAGraph()
s = torch.cuda.Stream()
s.wait_stream(torch.cuda.current_stream())
with torch.cuda.stream(s):
    run_eager(3)
torch.cuda.current_stream().wait_stream(s)
with torch.cuda.graph(g):
    run_eager(1000)

def t(fn, n=5):
    fn(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize()
    return (time.perf_counter()-t0)/n

te = t(lambda: run_eager(1000))
tg = t(lambda: g.replay())
print(f"eager 1000 tiny adds: {te*1e3:.2f} ms -> {te*1e3:.2f} us/kernel")
print(f"graph 1000 tiny adds: {tg*1e3:.2f} ms -> {tg*1e3:.2f} us/kernel")
# bigger grid kernel (2048 workgroups of trivial work)
big = torch.zeros(2048*256*4, device=dev)
def run_big(n):
    for _ in range(n): big.fill_(1.0)
g2 = torch.cuda.CUDAGraph()
s2 = torch.cuda.Stream(); s2.wait_stream(torch.cuda.current_stream())
with torch.cuda.stream(s2): run_big(3)
torch.cuda.current_stream().wait_stream(s2)
with torch.cuda.graph(g2): run_big(500)
tb = t(lambda: g2.replay())
print(f"graph 500 2MB fills: {tb*1e3:.2f} ms -> {tb*2:.2f} us/kernel")
