# -*- coding: utf-8 -*-
"""NHWC 3x3 convolution modules on the MFMA kernel pack (round 2).

Default GPU training path for the CIFAR ResNet body convs (Co == Ci,
(Co, W) in {(16,32), (32,16), (64,8)}) and the stride-2 transitions:

* forward: `hip/convfwd.h` conv3x3_bn_fwd_k — direct implicit GEMM with
  a fused BN-stats epilogue (the following BN skips its stats pass via
  the `_ft_bn_part` handoff) — 5.1-6.6 us/call;
* backward-data: conv3x3_dgrad_k (the forward's mirror) — 5.0-6.7 us
  vs MIOpen's 23-24;
* weight gradient: MFMA wrw v2 (`hip/convwrw2.h`) for C16 where it
  beats MIOpen-incl-wrappers (23.4 vs 33 us); MIOpen elsewhere;
* deferred BN backward (opt-in): the BN's elementwise dx pass runs
  inside the dgrad staging, handed off through a data_ptr-keyed side
  table (python tensor attrs do not survive the autograd engine's
  rewrapping).

Anything else falls back to stock F.conv2d autograd.  Per-call evidence:
profiles/r02_bench_notes.md; kernel details: hip/README.md.
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
# MFMA direct conv FORWARD with fused BN-stats epilogue (hip/convfwd.h):
# 5.1-6.6 us/call vs MIOpen's solver stack AND the following BN's stats
# pass dies (its partials come out of the conv epilogue).  Default ON.
_FWD_ENABLED = os.environ.get('FEDTORCH_MFMA_FWD', '1') == '1'
# MFMA wrw v2 (hip/convwrw2.h): transposed-LDS staging + alignbit tap
# shifts.  Its MAIN kernel (12.8-15.5 us/call) beats MIOpen's igemm wrw,
# but the partial-reduce/cast/zero auxiliary kernels per call still cost
# more than they save in-bench (148.8k vs 163.5k samples/s flagship), so
# default OFF until the partial pipeline is dieted.  FEDTORCH_WRW2=1 to
# enable.
_WRW2_ENABLED = os.environ.get('FEDTORCH_WRW2', '0') == '1'
# per-shape wrw2: the v2 kernel beats MIOpen-incl-wrappers at C16
# (23.4 vs 33 us) but not at C32/C64 — use it where it wins
_WRW2_SHAPES = set(
    int(c) for c in os.environ.get('FEDTORCH_WRW2_C', '16').split(',')
    if c.strip())
# MFMA direct conv backward-data (hip/convfwd.h conv3x3_dgrad_k): the fwd
# kernel's mirror, 5.0-6.7 us/call vs MIOpen's 23-24.  Default ON.
_DGRAD_ENABLED = os.environ.get('FEDTORCH_MFMA_DGRAD', '1') == '1'
# Deferred BN backward (hip/convfwd.h TRF path): the BN's elementwise dx
# pass runs inside the conv dgrad staging (which also writes the
# transformed dy for the wrw consumer) — the bnh_bwd_dx kernel leaves the
# hot path.  Handshake: _FusedBNFunction.backward registers
# (xbn, z, coefs, relu) under its outgoing grad tensor's data_ptr; the
# producing conv's backward pops it (python tensor attrs do not survive
# the autograd engine's rewrapping, a data_ptr-keyed side table does).
# MEASURED A WASH at the flagship config (174.0k vs 174.7k samples/s):
# the killed bnh_bwd_dx pass (~145 us/step) is consumed by the coef/mask
# side kernels + the 2 extra tensors the transform stages; default OFF,
# capability kept (tests pin deferred == eager gradients).
_BNDEFER_ENABLED = os.environ.get('FEDTORCH_BN_DEFER', '0') == '1'
_BNBWD_TAGS = {}


def bn_defer_active():
    return _DGRAD_ENABLED and _BNDEFER_ENABLED and not ops.FORCE_EAGER


def register_bn_defer(grad, xbn, z, coefs, relu, dres=None):
    _BNBWD_TAGS[grad.data_ptr()] = (xbn, z, coefs, relu, dres)
_EMPTY = {}


def _empty(dev):
    e = _EMPTY.get(dev)
    if e is None:
        e = _EMPTY[dev] = torch.empty(0, device=dev)
    return e


class _Conv3x3BNFn(torch.autograd.Function):
    """y, part = conv3x3(x, w) with the BN-stats partials [grid, Co, 2]
    produced by the conv epilogue (consumed by bn_fwd_train_part so the
    following BN skips its stats pass).  Backward: dgrad/wrw via the same
    paths as _Conv3x3Fn."""

    @staticmethod
    def forward(ctx, x, weight):
        grid = x.shape[0] * (x.shape[2] // 8) * \
            (2 if weight.shape[0] == 64 else 1)
        part = torch.empty(grid, weight.shape[0], 2, device=x.device)
        e = _empty(x.device)
        y = ops._C.conv3x3_bn_fwd(x, weight, part, e, e, e, False)
        ctx.save_for_backward(x, weight)
        ctx.mark_non_differentiable(part)
        # without this the engine materializes a ZERO gradient for `part`
        # every backward: 18 fill kernels/step (~60 us) in the profile
        ctx.set_materialize_grads(False)
        return y, part

    @staticmethod
    def backward(ctx, dy, _dpart):
        x, weight = ctx.saved_tensors
        # pop the deferred-BN tag by the INCOMING tensor's ptr before any
        # contiguous() copy could change it (a missed tag would silently
        # skip the BN backward transform)
        tag = _BNBWD_TAGS.pop(dy.data_ptr(), None) if _BNBWD_TAGS else None
        dy = dy.contiguous(memory_format=_CL)
        if tag is None and _BNBWD_TAGS:
            tag = _BNBWD_TAGS.pop(dy.data_ptr(), None)
        # a live tag means the BN already SKIPPED its dx pass: always
        # honor it (ignoring it would treat dz as dy_conv silently)
        if tag is not None:
            xbn, z, coefs, relu, dres = tag
            e = _empty(dy.device)
            dx, dyc = ops._C.conv3x3_dgrad_bn(
                dy, weight, xbn, z if z is not None else xbn, coefs, relu,
                dres if dres is not None else e)
            if not ctx.needs_input_grad[0]:
                dx = None
            if _WRW2_ENABLED:
                dw = ops._C.conv3x3_wrw2(dyc, x)
            else:
                dw = torch.ops.aten.convolution_backward(
                    dyc, x, weight, None, [1, 1], [1, 1], [1, 1], False,
                    [0, 0], 1, [False, True, False])[1]
            return dx, dw
        use_wrw2 = _WRW2_ENABLED or weight.shape[0] in _WRW2_SHAPES
        custom_dx = _DGRAD_ENABLED
        custom_dw = use_wrw2 or _ENABLED
        if custom_dx and custom_dw:
            dx = ops._C.conv3x3_dgrad(dy, weight) \
                if ctx.needs_input_grad[0] else None
            dw = ops._C.conv3x3_wrw2(dy, x) if use_wrw2 \
                else ops._C.conv3x3_wrw(dy, x)
        elif custom_dx:
            dx = ops._C.conv3x3_dgrad(dy, weight) \
                if ctx.needs_input_grad[0] else None
            dw = torch.ops.aten.convolution_backward(
                dy, x, weight, None, [1, 1], [1, 1], [1, 1], False, [0, 0],
                1, [False, True, False])[1]
        elif custom_dw:
            dx = None
            if ctx.needs_input_grad[0]:
                dx = torch.ops.aten.convolution_backward(
                    dy, x, weight, None, [1, 1], [1, 1], [1, 1], False,
                    [0, 0], 1, [True, False, False])[0]
            dw = ops._C.conv3x3_wrw2(dy, x) if _WRW2_ENABLED \
                else ops._C.conv3x3_wrw(dy, x)
        else:
            # ONE combined call (dgrad+wrw split into two was ~0.2 ms/step
            # slower over the 19 body convs)
            dx, dw, _ = torch.ops.aten.convolution_backward(
                dy, x, weight, None, [1, 1], [1, 1], [1, 1], False, [0, 0],
                1, [bool(ctx.needs_input_grad[0]), True, False])
        return dx, dw


class _Conv3x3S2BNFn(torch.autograd.Function):
    """stride-2 transition conv (16->32, 32->64) with the fused BN-stats
    epilogue; backward = ONE combined MIOpen call (these run once per
    step each — the fwd CK solver + the following BN's stats pass are
    the pool)."""

    @staticmethod
    def forward(ctx, x, weight):
        H_out = x.shape[2] // 2
        grid = x.shape[0] * (H_out // 8) * (weight.shape[0] // 32)
        part = torch.empty(grid, weight.shape[0], 2, device=x.device)
        y = ops._C.conv3x3s2_bn_fwd(x, weight, part)
        ctx.save_for_backward(x, weight)
        ctx.mark_non_differentiable(part)
        ctx.set_materialize_grads(False)
        return y, part

    @staticmethod
    def backward(ctx, dy, _dpart):
        x, weight = ctx.saved_tensors
        tag = _BNBWD_TAGS.pop(dy.data_ptr(), None) if _BNBWD_TAGS else None
        dy = dy.contiguous(memory_format=_CL)
        if tag is None and _BNBWD_TAGS:
            tag = _BNBWD_TAGS.pop(dy.data_ptr(), None)
        if tag is not None:
            # a deferred BN backward landed on the stride-2 conv (no TRF
            # kernel here): materialize dy_conv = A*mask(dz) + B + D*x_bn
            # eagerly (2 transition convs per step) and fill dres
            xbn, z, coefs, relu, dres = tag
            C = dy.shape[1]
            g = dy
            if relu and z is not None:
                g = dy * (z > 0)
            if dres is not None:
                dres.copy_(g)
            A = coefs[0].view(1, C, 1, 1)
            B = coefs[1].view(1, C, 1, 1)
            D = coefs[2].view(1, C, 1, 1)
            dy = (g.float() * A + B + D * xbn.float()).bfloat16() \
                .contiguous(memory_format=_CL)
        dx, dw, _ = torch.ops.aten.convolution_backward(
            dy, x, weight, None, [2, 2], [1, 1], [1, 1], False, [0, 0],
            1, [bool(ctx.needs_input_grad[0]), True, False])
        return dx, dw


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


class _Conv1x1S2Fn(torch.autograd.Function):
    """1x1 stride-2 downsample conv as three library GEMMs over the
    even-subsampled input (the ResNet transition shortcut): MIOpen's CK
    solvers for this shape carry batched-GEMM + wrapper kernels; a plain
    hipBLASLt GEMM over the strided slice is smaller and simpler.
    y[p, co] = sum_ci x_even[p, ci] w[co, ci]."""

    @staticmethod
    def forward(ctx, x, weight):
        N, Ci, H, W = x.shape
        Co = weight.shape[0]
        xe = x[:, :, ::2, ::2].contiguous(memory_format=_CL)
        wm = weight.reshape(Co, Ci)
        # channels_last [N,C,H,W] -> NHWC rows: permute view, flatten
        xr = xe.permute(0, 2, 3, 1).reshape(-1, Ci)
        y = (xr @ wm.t()).reshape(N, H // 2, W // 2, Co) \
            .permute(0, 3, 1, 2).contiguous(memory_format=_CL)
        ctx.save_for_backward(xe, weight)
        ctx.in_shape = (N, Ci, H, W)
        return y

    @staticmethod
    def backward(ctx, dy):
        xe, weight = ctx.saved_tensors
        N, Ci, H, W = ctx.in_shape
        Co = weight.shape[0]
        dy = dy.contiguous(memory_format=_CL)
        dyr = dy.permute(0, 2, 3, 1).reshape(-1, Co)
        dx = None
        if ctx.needs_input_grad[0]:
            dxe = (dyr @ weight.reshape(Co, Ci)) \
                .reshape(N, H // 2, W // 2, Ci).permute(0, 3, 1, 2)
            dx = torch.zeros(N, Ci, H, W, dtype=dy.dtype, device=dy.device
                             ).contiguous(memory_format=_CL)
            dx[:, :, ::2, ::2] = dxe
        xr = xe.permute(0, 2, 3, 1).reshape(-1, Ci)
        dw = (dyr.t() @ xr).reshape(Co, Ci, 1, 1) \
            .contiguous(memory_format=_CL)
        return dx, dw


_CONV1X1_GEMM = os.environ.get('FEDTORCH_CONV1X1_GEMM', '0') == '1'


class NhwcConv1x1S2(nn.Conv2d):
    """Drop-in 1x1/stride-2 bias-free Conv2d (the ResNet downsample
    shortcut).  The GEMM path (default OFF) measured SLOWER in-bench
    (160.0k vs 177.0k flagship): the K=16384 reduction wrw GEMM and the
    strided dx scatter cost more than MIOpen's CK solvers here."""

    def forward(self, x):
        w = self.weight
        use = (_CONV1X1_GEMM
               and self.training and x.is_cuda and x.dim() == 4
               and self.bias is None and self.stride == (2, 2)
               and self.kernel_size == (1, 1)
               and x.dtype == torch.bfloat16
               and x.is_contiguous(memory_format=_CL)
               and not ops.FORCE_EAGER)
        if use:
            wb = w if w.dtype == torch.bfloat16 else w.bfloat16()
            return _Conv1x1S2Fn.apply(x, wb)
        return F.conv2d(x, w, self.bias, self.stride, self.padding,
                        self.dilation, self.groups)


class NhwcConv3x3(nn.Conv2d):
    """Drop-in 3x3/s1/p1 bias-free Conv2d whose channels_last bf16 GPU path
    uses the MFMA wrw kernel.  state_dict layout is the stock Conv2d's."""

    def forward(self, x):
        w = self.weight
        use_fwd = (_FWD_ENABLED and self.training and x.is_cuda
                   and x.dim() == 4 and self.bias is None
                   and self.stride == (1, 1) and self.padding == (1, 1)
                   and x.dtype == torch.bfloat16
                   and self.in_channels == self.out_channels
                   and (self.out_channels, x.shape[3]) in _SHAPES
                   and x.shape[2] % 8 == 0
                   and x.is_contiguous(memory_format=_CL)
                   and ops.hip_available() and not ops.FORCE_EAGER)
        if use_fwd:
            wb = w if w.dtype == torch.bfloat16 else \
                w.bfloat16().contiguous(memory_format=_CL)
            if wb.is_contiguous(memory_format=_CL):
                y, part = _Conv3x3BNFn.apply(x, wb)
                y._ft_bn_part = part  # consumed by FusedBatchNorm2d
                return y
        use_s2 = (_FWD_ENABLED and self.training and x.is_cuda
                  and x.dim() == 4 and self.bias is None
                  and self.stride == (2, 2) and self.padding == (1, 1)
                  and x.dtype == torch.bfloat16
                  and self.out_channels == 2 * self.in_channels
                  and (self.in_channels, x.shape[3]) in ((16, 32), (32, 16))
                  and x.shape[2] % 16 == 0
                  and x.is_contiguous(memory_format=_CL)
                  and ops.hip_available() and not ops.FORCE_EAGER)
        if use_s2:
            wb = w if w.dtype == torch.bfloat16 else \
                w.bfloat16().contiguous(memory_format=_CL)
            if wb.is_contiguous(memory_format=_CL):
                y, part = _Conv3x3S2BNFn.apply(x, wb)
                y._ft_bn_part = part
                return y
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
