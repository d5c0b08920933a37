# -*- coding: utf-8 -*-
"""NHWC stem convolution module (gfx950 kernels, `hip/stemconv.h`).

MIOpen's NHWC bf16 solvers fall back to ``naive_conv_*`` for 3-input-channel
convolutions (the CIFAR ResNet stem), which is what forced the NCHW layout
and its per-conv ``batched_transpose`` wrapper kernels.  This module runs the
stem with hand-written channels_last kernels (3x3 / stride 1 / pad 1) so the
whole model can stay NHWC:

* forward reads the fp32 weight straight from the parameter arena (no
  per-step autocast cast kernel) and emits bf16 when autocast is active;
* backward computes only the weight gradient (the stem input is the data
  batch); falls back to eager conv2d whenever the input needs grad, the
  layout isn't channels_last, or shapes don't match.
"""
import torch
import torch.nn as nn
import torch.nn.functional as F

from fedtorch_amd import ops


class _StemConvFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, out_bf16):
        y = ops._C.stem_conv_fwd(x, weight, bool(out_bf16))
        ctx.save_for_backward(x)
        return y

    @staticmethod
    def backward(ctx, dy):
        (x,) = ctx.saved_tensors
        dw = ops._C.stem_conv_wrw(
            dy.contiguous(memory_format=torch.channels_last), x)
        return None, dw, None


class NhwcStemConv(nn.Conv2d):
    """Drop-in 3x3/s1/p1 Conv2d whose channels_last GPU path runs the custom
    stem kernels.  state_dict layout is that of the stock Conv2d."""

    def forward(self, x):
        use = (x.is_cuda and x.dim() == 4 and not x.requires_grad
               and self.bias is None and self.stride == (1, 1)
               and self.padding == (1, 1)
               and x.is_contiguous(memory_format=torch.channels_last)
               and self.weight.is_contiguous(
                   memory_format=torch.channels_last)
               and self.weight.dtype == torch.float32
               and self.in_channels == 3
               and self.out_channels in (16, 32)
               and ops.hip_available() and not ops.FORCE_EAGER)
        if not use:
            return F.conv2d(x, self.weight, self.bias, self.stride,
                            self.padding, self.dilation, self.groups)
        out_bf16 = torch.is_autocast_enabled('cuda')
        # run outside autocast: the kernel consumes fp32 weights directly
        with torch.autocast('cuda', enabled=False):
            return _StemConvFn.apply(x, self.weight, out_bf16)


def convert_stem(model):
    """Replace an eligible `conv1` stem with NhwcStemConv (weights reused in
    place; call BEFORE Arena construction)."""
    conv = getattr(model, 'conv1', None)
    if (isinstance(conv, nn.Conv2d) and not isinstance(conv, NhwcStemConv)
            and conv.kernel_size == (3, 3) and conv.stride == (1, 1)
            and conv.padding == (1, 1) and conv.bias is None
            and conv.in_channels == 3 and conv.out_channels in (16, 32)):
        new = NhwcStemConv(conv.in_channels, conv.out_channels, 3,
                           stride=1, padding=1, bias=False)
        new.weight = conv.weight
        model.conv1 = new
    return model
