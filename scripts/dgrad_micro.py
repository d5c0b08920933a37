import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import fedtorch_amd.ops as ops
cl = torch.channels_last
for C, W in [(16, 32), (32, 16), (64, 8)]:
    dy = torch.randn(256, C, W, W, device='cuda').bfloat16().contiguous(memory_format=cl)
    w = (torch.randn(C, C, 3, 3, device='cuda') / (3 * C) ** 0.5).bfloat16().contiguous(memory_format=cl)
    x = torch.randn(256, C, W, W, device='cuda').bfloat16().contiguous(memory_format=cl)
    dx = ops._C.conv3x3_dgrad(dy, w)
    ref = torch.ops.aten.convolution_backward(
        dy.float(), x.float(), w.float(), None, [1, 1], [1, 1], [1, 1],
        False, [0, 0], 1, [True, False, False])[0]
    rel = (dx.float() - ref).abs().max().item() / ref.abs().max().item()
    e0, e1 = torch.cuda.Event(True), torch.cuda.Event(True)
    for _ in range(20):
        ops._C.conv3x3_dgrad(dy, w)
    torch.cuda.synchronize(); e0.record()
    for _ in range(100):
        ops._C.conv3x3_dgrad(dy, w)
    e1.record(); torch.cuda.synchronize()
    mine = e0.elapsed_time(e1) * 10
    def mi():
        return torch.ops.aten.convolution_backward(
            dy, x, w, None, [1, 1], [1, 1], [1, 1], False, [0, 0], 1,
            [True, False, False])[0]
    for _ in range(20):
        mi()
    torch.cuda.synchronize(); e0.record()
    for _ in range(100):
        mi()
    e1.record(); torch.cuda.synchronize()
    mio = e0.elapsed_time(e1) * 10
    print('C=%-3d  dgrad mine %6.1f us (rel %.5f)  miopen %6.1f us'
          % (C, mine, rel, mio), flush=True)
    assert rel < 0.01, 'DGRAD FAIL'
print('DGRAD_OK')
