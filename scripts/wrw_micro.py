#!/usr/bin/env python3
"""Micro-timing of conv3x3_wrw vs MIOpen wrw per shape (CUDA events)."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import torch.nn.functional as F
import fedtorch_amd.ops as ops

cl = torch.channels_last
N = int(os.environ.get('WRW_N', '256'))
for C, W in [(16, 32), (32, 16), (64, 8)]:
    x = torch.randn(N, C, W, W, device='cuda').bfloat16().contiguous(memory_format=cl)
    dy = torch.randn(N, C, W, W, device='cuda').bfloat16().contiguous(memory_format=cl)
    w = torch.randn(C, C, 3, 3, device='cuda').bfloat16().contiguous(memory_format=cl)
    for _ in range(10):
        ops._C.conv3x3_wrw(dy, x)
    torch.cuda.synchronize()
    e0, e1 = torch.cuda.Event(True), torch.cuda.Event(True)
    e0.record()
    for _ in range(100):
        ops._C.conv3x3_wrw(dy, x)
    e1.record(); torch.cuda.synchronize()
    mine = e0.elapsed_time(e1) * 10
    # MIOpen wrw via aten
    def mi():
        return torch.ops.aten.convolution_backward(
            dy, x, w, None, [1, 1], [1, 1], [1, 1], False, [0, 0], 1,
            [False, True, False])[1]
    for _ in range(10):
        mi()
    torch.cuda.synchronize()
    e0.record()
    for _ in range(100):
        mi()
    e1.record(); torch.cuda.synchronize()
    miopen = e0.elapsed_time(e1) * 10
    # v2 correctness vs fp32 reference + timing
    dw2 = ops._C.conv3x3_wrw2(dy, x)
    ref = torch.ops.aten.convolution_backward(
        dy.float(), x.float(), w.float(), None, [1, 1], [1, 1], [1, 1],
        False, [0, 0], 1, [False, True, False])[1]
    rel = (dw2.float() - ref).abs().max().item() / ref.abs().max().item()
    for _ in range(10):
        ops._C.conv3x3_wrw2(dy, x)
    torch.cuda.synchronize()
    e0.record()
    for _ in range(100):
        ops._C.conv3x3_wrw2(dy, x)
    e1.record(); torch.cuda.synchronize()
    v2 = e0.elapsed_time(e1) * 10
    print('C=%-3d W=%-3d N=%d  v1 %7.1f us  v2 %7.1f us (rel err %.5f)  miopen %7.1f us'
          % (C, W, N, mine, v2, rel, miopen), flush=True)
    assert rel < 0.01, 'WRW2 numerics FAIL'
