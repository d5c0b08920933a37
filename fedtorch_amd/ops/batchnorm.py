# -*- coding: utf-8 -*-
"""Fused spatial BatchNorm module (gfx950 kernels, `hip/batchnorm.h`).

Training fwd/bwd run 2+2 hand-written streaming kernels instead of
MIOpen's 3+3 (~40 % of ResNet-20 step kernel time, see
profiles/r01_bench_notes.md), stats in fp32, I/O in the input dtype (bf16
under autocast — no cast passes).  Eval mode and CPU fall back to the
stock nn.BatchNorm2d path; state_dict layout is unchanged.
"""
import torch
import torch.nn as nn

from fedtorch_amd import ops


class _FusedBNFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, running_mean, running_var, momentum,
                eps):
        empty = torch.empty(0, device=x.device)
        y, save_mean, save_ivar = ops._C.bn_fwd_train(
            x, weight if weight is not None else empty,
            bias if bias is not None else empty,
            running_mean if running_mean is not None else empty,
            running_var if running_var is not None else empty,
            float(eps), float(momentum), False)
        ctx.save_for_backward(x, weight, save_mean, save_ivar)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, save_mean, save_ivar = ctx.saved_tensors
        empty = torch.empty(0, device=x.device)
        dx, dweight, dbias = ops._C.bn_bwd(
            dy.contiguous(), x, empty, save_mean, save_ivar,
            weight if weight is not None else empty, False)
        return (dx, dweight if weight is not None else None,
                dbias if weight is not None else None, None, None, None,
                None)


class FusedBatchNorm2d(nn.BatchNorm2d):
    """Drop-in BatchNorm2d: fused HIP kernels for GPU training, stock path
    otherwise."""

    def forward(self, x):
        use_fused = (self.training and x.is_cuda and x.dim() == 4
                     and ops.hip_available() and not ops.FORCE_EAGER)
        if not use_fused:
            return super().forward(x)
        if self.track_running_stats and self.num_batches_tracked is not None:
            self.num_batches_tracked.add_(1)
        momentum = self.momentum if self.momentum is not None else 0.1
        return _FusedBNFunction.apply(
            x.contiguous(), self.weight, self.bias,
            self.running_mean if self.track_running_stats else None,
            self.running_var if self.track_running_stats else None,
            momentum, self.eps)


def convert_to_fused_bn(module):
    """Recursively replace nn.BatchNorm2d with FusedBatchNorm2d (params and
    buffers are re-used in place, state_dict layout unchanged)."""
    for name, child in module.named_children():
        if type(child) is nn.BatchNorm2d:
            fused = FusedBatchNorm2d(
                child.num_features, eps=child.eps, momentum=child.momentum,
                affine=child.affine,
                track_running_stats=child.track_running_stats)
            fused.weight = child.weight
            fused.bias = child.bias
            if child.track_running_stats:
                fused.running_mean = child.running_mean
                fused.running_var = child.running_var
                fused.num_batches_tracked = child.num_batches_tracked
            setattr(module, name, fused)
        else:
            convert_to_fused_bn(child)
    return module
