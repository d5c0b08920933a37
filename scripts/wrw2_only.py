import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import fedtorch_amd.ops as ops
cl = torch.channels_last
for C, W in [(16, 32), (32, 16), (64, 8)]:
    x = torch.randn(256, C, W, W, device='cuda').bfloat16().contiguous(memory_format=cl)
    dy = torch.randn(256, C, W, W, device='cuda').bfloat16().contiguous(memory_format=cl)
    for _ in range(110):
        ops._C.conv3x3_wrw2(dy, x)
    torch.cuda.synchronize()
