# -*- coding: utf-8 -*-
"""Flat contiguous parameter arena — the central MI355X-native data structure.

The reference iterates ``model.parameters()`` in Python for every hot path
(aggregation `comms/algorithms/federated/fedavg.py:30-94`, SGD step
`components/optimizers/sgd.py:81-128`, distribution `federated/misc.py:26`),
which costs P ~ 65 tiny kernel launches + P tiny messages per model per sync.
Here every replica's parameters live in ONE contiguous fp32 buffer:

* every optimizer / aggregation hot path is ONE fused HIP kernel launch;
* every RCCL collective moves ONE arena-sized message over xGMI;
* auxiliary state (server model, control variates, FedGATE delta/memory,
  momentum) are plain flat tensors of the same size — no nn.Module clones
  (the reference deep-copies whole modules for these, `nodes/nodes.py:87-112`).

Parameter tensors are re-pointed (``p.data = view``) into the arena, and
``p.grad`` is pointed into a twin gradient arena, so autograd accumulates
directly into the flat buffer — no flatten/unflatten copies anywhere.
Offsets are aligned to 64 elements (256 B) so vectorized (float4 / short8)
kernel loads stay aligned per parameter.
"""
import torch

ALIGN = 64  # elements; 256 B for fp32 — one LDS bank row on gfx950


def _aligned(n, align=ALIGN):
    return (n + align - 1) // align * align


class Arena(object):
    """Flat fp32 arena backing the parameters (and grads) of a module."""

    def __init__(self, module, device=None, dtype=torch.float32,
                 with_grads=True, no_decay_predicate=None):
        """``no_decay_predicate(name, param) -> bool`` marks parameters that
        weight decay must skip (the reference zeroes wd for params whose name
        contains 'bn', `components/optimizer.py:8-16`).  Those are packed at
        the END of the arena so the fused SGD kernel applies wd to the flat
        prefix ``[0, wd_numel)`` with zero extra memory traffic."""
        self.module = module
        named = [(n, p) for n, p in module.named_parameters()
                 if p.requires_grad]
        if len(named) == 0:
            raise ValueError('module has no trainable parameters')
        if no_decay_predicate is None:
            no_decay_predicate = lambda name, p: 'bn' in name  # noqa: E731
        decay = [(n, p) for n, p in named if not no_decay_predicate(n, p)]
        nodecay = [(n, p) for n, p in named if no_decay_predicate(n, p)]
        ordered = decay + nodecay
        self.names = [n for n, _ in ordered]
        params = [p for _, p in ordered]
        self.dtype = dtype
        self.device = device if device is not None else params[0].device
        self.offsets, self.numels, self.shapes = [], [], []
        # channels_last 4D params are packed in NHWC element order so the
        # arena view handed back to the module keeps channels_last strides
        # (the end-to-end NHWC path needs conv weights in that format; every
        # arena op is elementwise over the flat buffer, so order is free).
        self.cl = []
        total = 0
        for p in params:
            self.offsets.append(total)
            self.numels.append(p.numel())
            self.shapes.append(p.shape)
            self.cl.append(
                p.dim() == 4
                and p.data.is_contiguous(memory_format=torch.channels_last)
                and not p.data.is_contiguous())
            total += _aligned(p.numel())
        # wd applies to flat[:wd_numel] (includes pad gaps, which stay zero).
        self.wd_numel = (self.offsets[len(decay) - 1] +
                         _aligned(self.numels[len(decay) - 1])
                         if decay else 0)
        if len(nodecay) == 0:
            self.wd_numel = total
        self.numel = total
        # NOTE: the pad gaps stay zero forever: every fused op is linear in
        # the buffers, so zero gaps stay zero through steps/collectives.
        self.flat = torch.zeros(total, dtype=dtype, device=self.device)
        for p, off, n, cl in zip(params, self.offsets, self.numels, self.cl):
            if cl:
                nhwc = p.data.permute(0, 2, 3, 1)
                self.flat[off:off + n].copy_(nhwc.reshape(-1).to(dtype))
                p.data = self._cl_view(self.flat, off, n, p.shape)
            else:
                self.flat[off:off + n].copy_(p.data.reshape(-1).to(dtype))
                p.data = self.flat[off:off + n].view(p.shape)
        self.params = params
        self.grad = None
        if with_grads:
            self.grad = torch.zeros_like(self.flat)
            self.attach_grads()
        # BatchNorm running stats live in a secondary flat buffer so sync
        # can average them (the reference never aggregates buffers — a
        # BN-model server would keep INIT stats and its eval-mode outputs
        # explode; its centered experiments only used BN-free models or
        # track_running_stats=False, `models/nonconvex/mlp.py:25`).
        self.buf_flat = None
        self._pack_running_stats(module)

    def _pack_running_stats(self, module):
        entries = []
        total = 0
        for m in module.modules():
            for bname in ('running_mean', 'running_var'):
                b = getattr(m, '_buffers', {}).get(bname, None)
                if b is not None and b.is_floating_point():
                    entries.append((m, bname, total, b.numel()))
                    total += _aligned(b.numel())
        if not entries:
            return
        self.buf_flat = torch.zeros(total, dtype=self.dtype,
                                    device=self.device)
        for m, bname, off, n in entries:
            self.buf_flat[off:off + n].copy_(m._buffers[bname].reshape(-1))
            m._buffers[bname] = self.buf_flat[off:off + n]

    @staticmethod
    def _cl_view(flat, off, n, shape):
        """channels_last view of an arena region storing NHWC element order."""
        N, C, H, W = shape
        return flat[off:off + n].view(N, H, W, C).permute(0, 3, 1, 2)

    # ---- bf16 compute copy (fp32 master) -----------------------------------
    # Conv/Linear weights are what autocast casts to bf16 EVERY step (~48
    # cast kernels per ResNet-20 step under hipGraph replay).  Instead, those
    # parameters point at a bf16 twin of the arena; the fused SGD kernel
    # refreshes the twin in its own pass over the fp32 master, and the grad
    # gather casts the (now bf16) stolen grads back to fp32 during its copy.
    # Intended for the hipGraph/stolen-grad path (autograd requires p.grad
    # dtype == p dtype, so attached-view mode keeps fp32 params).
    half_flat = None

    def enable_bf16_compute(self):
        """Re-point conv (4D) and nn.Linear parameters at a bf16 twin
        arena.  Returns self."""
        if self.half_flat is not None:
            return self
        import torch.nn as nn
        linear_params = set()
        for m in self.module.modules():
            if isinstance(m, nn.Linear):
                for p in m.parameters(recurse=False):
                    linear_params.add(id(p))
        self.half_flat = torch.zeros(self.numel, dtype=torch.bfloat16,
                                     device=self.device)
        self.half_flat.copy_(self.flat)
        self.bf16 = []
        for p, off, n, cl in zip(self.params, self.offsets, self.numels,
                                 self.cl):
            use = p.dim() == 4 or id(p) in linear_params
            self.bf16.append(use)
            if not use:
                continue
            if cl:
                p.data = self._cl_view(self.half_flat, off, n, p.shape)
            else:
                p.data = self.half_flat[off:off + n].view(p.shape)
        self._gather_state = None
        return self

    def disable_bf16_compute(self):
        """Restore fp32 master views (e.g. when falling back to eager)."""
        if self.half_flat is None:
            return
        for p, off, n, cl in zip(self.params, self.offsets, self.numels,
                                 self.cl):
            if cl:
                p.data = self._cl_view(self.flat, off, n, p.shape)
            else:
                p.data = self.flat[off:off + n].view(p.shape)
        self.half_flat = None
        self.bf16 = None
        self._gather_state = None

    def sync_half(self):
        """Refresh the bf16 twin after a direct mutation of the fp32 master
        (aggregation writes outside the fused SGD kernel)."""
        if self.half_flat is None:
            return
        from fedtorch_amd import ops
        if ops.hip_available() and self.flat.is_cuda:
            ops._C.cast_to_half(self.flat, self.half_flat)
        else:
            self.half_flat.copy_(self.flat)

    # ---- gradient plumbing -------------------------------------------------
    def attach_grads(self):
        """Point every p.grad at its view of the grad arena."""
        for p, off, n, cl in zip(self.params, self.offsets, self.numels,
                                 self.cl):
            if cl:
                p.grad = self._cl_view(self.grad, off, n, p.shape)
            else:
                p.grad = self.grad[off:off + n].view(p.shape)
        self._gather_state = None

    def zero_grad(self):
        self.grad.zero_()

    # ---- stolen-grad gather (hipGraph path) --------------------------------
    # With p.grad = None at capture time, AccumulateGrad STEALS the produced
    # tensor (no per-parameter fp32 add kernel); gather_grads() then runs ONE
    # HIP kernel copying every stolen buffer into the contiguous grad arena
    # the fused SGD / trackers read.  detach_grads() before capture,
    # gather_grads() after loss.backward() inside the capture.
    _GATHER_CHUNK = 4096  # fp32 elements per block

    def detach_grads(self):
        for p in self.params:
            p.grad = None
        self._gather_state = None

    def _build_gather_state(self):
        srcs, table = [], []
        for i, (p, off, n, cl) in enumerate(zip(self.params, self.offsets,
                                                self.numels, self.cl)):
            g = p.grad
            # a channels_last stolen grad's linear element order matches the
            # NHWC-packed arena region, so its flat copy is still direct
            dense = (g is not None
                     and (g.is_contiguous(
                             memory_format=torch.channels_last)
                          if cl else g.is_contiguous()))
            if not dense or g.dtype not in (torch.float32, torch.bfloat16):
                return None  # fall back to per-tensor copies
            bf16 = 1 if g.dtype == torch.bfloat16 else 0
            srcs.append(g.data_ptr())
            for c in range(0, n, self._GATHER_CHUNK):
                table.append((i, c, off + c, min(self._GATHER_CHUNK, n - c),
                              bf16))
        dev = self.flat.device
        return (torch.tensor(srcs, dtype=torch.int64, device=dev),
                torch.tensor(table, dtype=torch.int32, device=dev))

    def gather_grads(self):
        """One-kernel copy of the stolen per-parameter grads into the grad
        arena.  Builds the chunk table on first use (static shapes).

        hipGraph captures BAKE the current table tensors' addresses into
        the recorded kernel: the capturing code must keep a reference to
        `_gather_state` for the graph's lifetime (see
        trainings/graphstep.py `_keep`) — a later detach_grads() drops it
        here and the memory can be reused."""
        if getattr(self, '_gather_state', None) is None:
            from fedtorch_amd import ops
            state = None
            if self.flat.is_cuda and ops.hip_available():
                state = self._build_gather_state()
            self._gather_state = state if state is not None else ()
        if self._gather_state:
            from fedtorch_amd import ops
            ops._C.gather_grads(self._gather_state[0],
                                self._gather_state[1], self.grad)
        else:  # non-contiguous / exotic-dtype stolen grads: plain copies
            for p, off, n, cl in zip(self.params, self.offsets, self.numels,
                                     self.cl):
                g = p.grad.permute(0, 2, 3, 1) if cl else p.grad
                self.grad[off:off + n].copy_(g.reshape(-1).float())

    # ---- flat state helpers ------------------------------------------------
    def new_buffer(self, zero=True):
        """A detached flat tensor of arena shape (server copy, control
        variate, delta, memory, momentum, ...)."""
        return torch.zeros_like(self.flat) if zero else torch.empty_like(self.flat)

    def clone_flat(self):
        return self.flat.detach().clone()

    def load_flat(self, buf):
        """Set model parameters from a flat buffer (single device copy)."""
        self.flat.copy_(buf)

    def views_of(self, buf):
        """Per-parameter shaped views of any arena-shaped flat buffer."""
        return [buf[off:off + n].view(shape)
                for off, n, shape in zip(self.offsets, self.numels, self.shapes)]

    def check_views(self):
        """True if every param still points into the arena — or, in bf16
        compute mode, into the bf16 twin (a module-level ``p.data = ...``
        assignment elsewhere would silently detach it)."""
        ranges = [(self.flat.data_ptr(),
                   self.flat.data_ptr()
                   + self.flat.numel() * self.flat.element_size())]
        if self.half_flat is not None:
            ranges.append((self.half_flat.data_ptr(),
                           self.half_flat.data_ptr()
                           + self.half_flat.numel()
                           * self.half_flat.element_size()))
        return all(any(lo <= p.data_ptr() < hi for lo, hi in ranges)
                   for p in self.params)

    # ---- interop -----------------------------------------------------------
    def state_dict_flat(self):
        return {'flat': self.flat}

    def __repr__(self):
        return 'Arena(n_params={}, numel={}, device={})'.format(
            len(self.params), self.numel, self.flat.device)
