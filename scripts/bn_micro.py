#!/usr/bin/env python3
"""Microbenchmark of the NHWC fused-BN kernel pack on the three ResNet-20
layer shapes (b256). Grid knobs come from FT_BNH_* env vars (read once per
process), so the sweep driver re-execs itself per config."""
import os
import subprocess
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

SHAPES = [(256, 16, 32, 32), (256, 32, 16, 16), (256, 64, 8, 8)]


def measure():
    import torch
    from fedtorch_amd.ops.batchnorm import BNAddReLU, convert_to_fused_bn
    cl = torch.channels_last
    rows = []
    for (N, C, H, W) in SHAPES:
        x = torch.randn(N, C, H, W, device='cuda',
                        dtype=torch.bfloat16).contiguous(memory_format=cl)
        res = torch.randn_like(x).contiguous(memory_format=cl)
        mod = convert_to_fused_bn(BNAddReLU(C).cuda())
        xr = x.clone().requires_grad_(True)
        rr = res.clone().requires_grad_(True)
        g = torch.randn_like(x)

        def one():
            y = mod(xr, rr)
            y.backward(g)
            xr.grad = None
            rr.grad = None

        for _ in range(20):
            one()
        torch.cuda.synchronize()
        e0 = torch.cuda.Event(True)
        e1 = torch.cuda.Event(True)
        e0.record()
        iters = 200
        for _ in range(iters):
            one()
        e1.record()
        torch.cuda.synchronize()
        us = e0.elapsed_time(e1) * 1000 / iters
        rows.append('C=%-3d fwd+bwd %.1f us' % (C, us))
    print(' | '.join(rows), flush=True)


if __name__ == '__main__':
    if len(sys.argv) > 1 and sys.argv[1] == 'one':
        measure()
        raise SystemExit(0)
    combos = [
        dict(),  # defaults
        dict(FT_BNH_RED_CAP='64'),
        dict(FT_BNH_RED_CAP='256'),
        dict(FT_BNH_RED_CAP='256', FT_BNH_RED_ITERS='4'),
        dict(FT_BNH_EW_CAP='512', FT_BNH_EW_ITERS='4'),
        dict(FT_BNH_EW_CAP='2048', FT_BNH_EW_ITERS='1'),
        dict(FT_BNH_RED_CAP='256', FT_BNH_EW_CAP='2048',
             FT_BNH_EW_ITERS='1'),
    ]
    for c in combos:
        env = dict(os.environ, **c)
        label = ','.join('%s=%s' % kv for kv in c.items()) or 'defaults'
        print('== %s' % label, flush=True)
        subprocess.run([sys.executable, __file__, 'one'], env=env, check=True)
