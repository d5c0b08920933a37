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

_VN = {torch.bfloat16: 8, torch.float32: 4, torch.float64: 2}


def _is_cl(x):
    """channels_last-contiguous 4D (and not plainly contiguous)."""
    return (x.dim() == 4
            and x.is_contiguous(memory_format=torch.channels_last)
            and not x.is_contiguous())


def _nhwc_ok(C, dtype):
    """Eligibility of the NHWC kernel path (ops/hip/batchnorm.h bnh_*):
    C a multiple of the 16 B vector width with a power-of-two group count
    ≤ 64 and C ≤ 256."""
    vn = _VN.get(dtype)
    if vn is None or C % vn or C > 256:
        return False
    cgc = C // vn
    return cgc <= 64 and (cgc & (cgc - 1)) == 0


class _FusedBNFunction(torch.autograd.Function):
    """y = [relu](bn(x) [+ res]); res fuses a ResNet residual add into the
    normalize kernel (and its gradient — the ReLU-masked dy — into the dx
    kernel)."""

    @staticmethod
    def forward(ctx, x, weight, bias, running_mean, running_var, momentum,
                eps, relu, res, part=None):
        empty = torch.empty(0, device=x.device)
        if part is not None:
            # stats partials already produced by the conv3x3_bn_fwd
            # epilogue: ONE norm launch, no stats pass
            y, save_mean, save_ivar, mask = ops._C.bn_fwd_train_part(
                x, part, weight if weight is not None else empty,
                bias if bias is not None else empty,
                running_mean if running_mean is not None else empty,
                running_var if running_var is not None else empty,
                float(eps), float(momentum), bool(relu),
                res if res is not None else empty)
        else:
            y, save_mean, save_ivar, mask = ops._C.bn_fwd_train(
                x, weight if weight is not None else empty,
                bias if bias is not None else empty,
                running_mean if running_mean is not None else empty,
                running_var if running_var is not None else empty,
                float(eps), float(momentum), bool(relu),
                res if res is not None else empty)
        ctx.relu = bool(relu)
        ctx.has_res = res is not None
        ctx.nhwc = _is_cl(x)
        ctx.has_mask = mask.numel() > 0
        ctx.from_fused_conv = part is not None
        if ctx.has_mask:
            # ReLU bitmask replaces the saved output (bwd reads 1 bit/elem)
            ctx.save_for_backward(x, weight, save_mean, save_ivar, mask)
        elif relu:
            ctx.save_for_backward(x, weight, save_mean, save_ivar, y)
        else:
            ctx.save_for_backward(x, weight, save_mean, save_ivar)
        return y

    @staticmethod
    def backward(ctx, dy):
        mask = None
        if ctx.has_mask:
            x, weight, save_mean, save_ivar, mask = ctx.saved_tensors
            y = torch.empty(0, device=x.device)
        elif ctx.relu:
            x, weight, save_mean, save_ivar, y = ctx.saved_tensors
        else:
            x, weight, save_mean, save_ivar = ctx.saved_tensors
            y = torch.empty(0, device=x.device)
        if ctx.nhwc:
            dy = dy.contiguous(memory_format=torch.channels_last)
        else:
            dy = dy.contiguous()
        empty = torch.empty(0, device=x.device)
        from fedtorch_amd.ops import conv3x3 as _c3
        if (ctx.from_fused_conv and ctx.nhwc and not ctx.has_mask
                and _c3.bn_defer_active()):
            # deferred dx: the producing conv's backward applies the
            # per-channel affine transform while staging (and emits the
            # transformed dy for its wrw) — return dz tagged via the
            # side table instead of running bnh_bwd_dx here
            z = y if ctx.relu else None
            # dres is ALLOCATED here but FILLED by the consuming conv's
            # staging kernel (stream order guarantees the fill lands
            # before any accumulation reads it); bn_bwd_defer only runs
            # its mask kernel when the conv cannot fill (defensive: the
            # registry consumer always can on this path)
            coefs, dweight, dbias, _ = ops._C.bn_bwd_defer(
                dy, x, z if z is not None else empty,
                save_mean, save_ivar,
                weight if weight is not None else empty,
                ctx.relu, False)
            dres = torch.empty_like(dy) if ctx.has_res else None
            _c3.register_bn_defer(dy, x, z, coefs, ctx.relu, dres)
            return (dy, dweight if weight is not None else None,
                    dbias if weight is not None else None, None, None,
                    None, None, None,
                    dres if ctx.has_res else None, None)
        dx, dweight, dbias, dres = ops._C.bn_bwd(
            dy, x, y, mask if mask is not None else
            torch.empty(0, device=x.device, dtype=torch.uint8),
            save_mean, save_ivar,
            weight if weight is not None else empty, ctx.relu, ctx.has_res)
        return (dx, dweight if weight is not None else None,
                dbias if weight is not None else None, None, None, None,
                None, None, dres if ctx.has_res else None, None)


class FusedBatchNorm2d(nn.BatchNorm2d):
    """Drop-in BatchNorm2d: fused HIP kernels for GPU training, stock path
    otherwise.  ``fuse_relu=True`` folds the following ReLU into the same
    kernels (fwd clamp + bwd mask on the saved output)."""

    fuse_relu = False
    _nbt_pending = 0  # lazy num_batches_tracked increments (flushed on save)

    def forward(self, x, res=None):
        cl = _is_cl(x)
        use_fused = (self.training and x.is_cuda and x.dim() == 4
                     and ops.hip_available() and not ops.FORCE_EAGER
                     and (not cl or _nhwc_ok(self.num_features, x.dtype)))
        if not use_fused:
            self._flush_nbt()
            y = super().forward(x)
            if res is not None:
                return torch.relu(y + res)
            return torch.relu(y) if self.fuse_relu else y
        if self.track_running_stats and self.num_batches_tracked is not None:
            # a GPU .add_(1) here is one extra kernel per BN layer per step
            # (it shows in rocprof as CUDAFunctorOnSelf_add<long>); count on
            # the host instead and flush into the buffer only when the
            # state_dict is read (checkpoint / sync).
            self._nbt_pending += 1
        momentum = self.momentum if self.momentum is not None else 0.1
        if cl:
            xc = x  # already channels_last-contiguous
            rc = (res.contiguous(memory_format=torch.channels_last)
                  if res is not None else None)
        else:
            xc = x.contiguous()
            rc = res.contiguous() if res is not None else None
        part = getattr(x, '_ft_bn_part', None) if cl else None
        if part is not None and not _nhwc_ok(self.num_features, x.dtype):
            part = None
        return _FusedBNFunction.apply(
            xc, self.weight, self.bias,
            self.running_mean if self.track_running_stats else None,
            self.running_var if self.track_running_stats else None,
            momentum, self.eps,
            self.fuse_relu or res is not None, rc, part)

    def _flush_nbt(self):
        if self._nbt_pending and self.num_batches_tracked is not None:
            self.num_batches_tracked.add_(self._nbt_pending)
            self._nbt_pending = 0

    def _save_to_state_dict(self, destination, prefix, keep_vars):
        self._flush_nbt()
        super()._save_to_state_dict(destination, prefix, keep_vars)


class BNReLU(nn.Module):
    """BatchNorm2d + ReLU as one module so the fused-BN conversion can fold
    the ReLU into the BN kernels (`convert_to_fused_bn`)."""

    def __init__(self, planes):
        super().__init__()
        self.bn = nn.BatchNorm2d(planes)

    def forward(self, x):
        y = self.bn(x)
        # when the inner BN is the fused kernel with fuse_relu, the clamp
        # already happened (checking the flag here keeps the module
        # deepcopy-safe — no bound-method monkeypatching).
        if getattr(self.bn, 'fuse_relu', False):
            return y
        return torch.relu(y)


class BNAddReLU(nn.Module):
    """``relu(bn(x) + res)`` — the ResNet block tail — as one module so the
    fused-BN conversion can fold the residual add AND the ReLU into the BN
    normalize/dx kernels (3 fewer kernels per block per step)."""

    def __init__(self, planes):
        super().__init__()
        self.bn = nn.BatchNorm2d(planes)

    def forward(self, x, res):
        if isinstance(self.bn, FusedBatchNorm2d):
            return self.bn(x, res=res)
        return torch.relu(self.bn(x) + res)


def convert_to_fused_bn(module):
    """Recursively replace nn.BatchNorm2d with FusedBatchNorm2d (params and
    buffers are re-used in place, state_dict layout unchanged)."""
    def _fused_of(child, relu):
        fused = FusedBatchNorm2d(
            child.num_features, eps=child.eps, momentum=child.momentum,
            affine=child.affine,
            track_running_stats=child.track_running_stats)
        fused.fuse_relu = relu
        fused.weight = child.weight
        fused.bias = child.bias
        if child.track_running_stats:
            fused.running_mean = child.running_mean
            fused.running_var = child.running_var
            fused.num_batches_tracked = child.num_batches_tracked
        return fused

    for name, child in module.named_children():
        if type(child) is nn.BatchNorm2d:
            setattr(module, name, _fused_of(child, relu=False))
        elif type(child) is BNReLU:
            child.bn = _fused_of(child.bn, relu=True)
        elif type(child) is BNAddReLU:
            # relu comes from the res= path at call time, not the flag
            child.bn = _fused_of(child.bn, relu=False)
        else:
            convert_to_fused_bn(child)
    return module
