import sys, os
sys.path.insert(0, '/root/repo')
import torch
from murmura_amd.ops import _load_ext
ext = _load_ext()
n_,c,h,w,k = 64,64,32,32,64
x = torch.randn(n_,c,h,w,device="cuda",dtype=torch.bfloat16).contiguous(memory_format=torch.channels_last)
dy = torch.randn(n_,k,h,w,device="cuda",dtype=torch.bfloat16).contiguous(memory_format=torch.channels_last)
for _ in range(20):
    ext.conv3x3s1_wrw(x, dy)
torch.cuda.synchronize()
