# -*- coding: utf-8 -*-
"""NHWC 3x3/s1/p1 conv with MFMA weight-gradient (`hip/convwrw.h`).

Forward and the data gradient stay on MIOpen's native-NHWC igemm solvers
(already fast); the WEIGHT gradient — whose MIOpen solvers carry
SubTensorOp workspace-zero and fp32→bf16 cast wrapper kernels (~280 us of
a 1.65 ms ResNet-20/b256 step, profiles/r01_bench_notes.md) — runs a
hand-written MFMA 16x16x32 kernel computing all 9 taps as tile-GEMMs over
the flattened position axis, no workspace, bf16 out.

Eligible shapes (compiled template instances): Co == Ci with
(Co, W) in {(16,32), (32,16), (64,8)} — the CIFAR ResNet body convs.
Anything else falls back to stock F.conv2d autograd.
"""
import os

import torch
import torch.nn as nn
import torch.nn.functional as F

from fedtorch_amd import ops

_CL = torch.channels_last
_SHAPES = {(16, 32), (32, 16), (64, 8)}
# Default OFF: the kernel is numerically exact (test_conv3x3_wrw_*) but at
# CIFAR sizes it measured 27-79 us/call vs MIOpen's 22-33 us (the LDS-staged
# tile pipeline is latency-bound at these tiny K-tiles; see
# profiles/r01_bench_notes.md "MFMA wrw" entry).  FEDTORCH_MFMA_WRW=1
# re-enables for experimentation.
_ENABLED = os.environ.get('FEDTORCH_MFMA_WRW', '0') == '1'


class _Conv3x3Fn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight):
        y = F.conv2d(x, weight, None, (1, 1), (1, 1))
        ctx.save_for_backward(x, weight)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight = ctx.saved_tensors
        dy = dy.contiguous(memory_format=_CL)
        dx = None
        if ctx.needs_input_grad[0]:
            dx = torch.ops.aten.convolution_backward(
                dy, x, weight, None, [1, 1], [1, 1], [1, 1], False, [0, 0],
                1, [True, False, False])[0]
        dw = ops._C.conv3x3_wrw(dy, x)
        return dx, dw


class NhwcConv3x3(nn.Conv2d):
    """Drop-in 3x3/s1/p1 bias-free Conv2d whose channels_last bf16 GPU path
    uses the MFMA wrw kernel.  state_dict layout is the stock Conv2d's."""

    def forward(self, x):
        w = self.weight
        use = (_ENABLED and torch.is_grad_enabled() and self.training
               and x.is_cuda
               and x.dim() == 4 and self.bias is None
               and self.stride == (1, 1) and self.padding == (1, 1)
               and x.dtype == torch.bfloat16 and w.dtype == torch.bfloat16
               and self.in_channels == self.out_channels
               and (self.out_channels, x.shape[3]) in _SHAPES
               and x.shape[2] % (32 // x.shape[3]) == 0
               and x.is_contiguous(memory_format=_CL)
               and w.is_contiguous(memory_format=_CL)
               and ops.hip_available() and not ops.FORCE_EAGER)
        if not use:
            return F.conv2d(x, w, self.bias, self.stride, self.padding,
                            self.dilation, self.groups)
        return _Conv3x3Fn.apply(x, w)
