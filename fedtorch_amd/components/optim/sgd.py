# -*- coding: utf-8 -*-
"""Dual-mode fused SGD over a flat arena.

The reference's custom SGD (`components/optimizers/sgd.py:67-128`) is the
single most-reused math primitive: a *local* step applies lr + "in" momentum
+ weight decay, the *sync* step is called with ``apply_lr=False,
scale=lr_scale_at_sync`` (so ``p -= scale * aggregated_diff``) with a separate
"out" momentum buffer.  It loops over P parameter tensors; here the whole
model is one fused kernel launch over the arena
(`fedtorch_amd/ops/fused_sgd.hip`).

Per-algorithm gradient corrections that the reference applies by mutating
``p.grad`` in the training loop (`comms/trainings/federated/main.py:116-129`)
are fused into the same kernel: set `prox` / `scaffold` / `fedgate` state via
:meth:`set_correction` before local steps.
"""
import torch

from fedtorch_amd import ops


class FusedSGD(object):
    def __init__(self, arena, lr, in_momentum=0.0, out_momentum=0.0,
                 dampening=0.0, weight_decay=0.0, nesterov=False):
        if nesterov and (in_momentum <= 0 or dampening != 0):
            raise ValueError('Nesterov momentum requires a momentum and zero '
                             'dampening')
        self.arena = arena
        # single param group; `adjust_learning_rate` writes group['lr'] every
        # step like the reference (`components/scheduler.py:9-29`).
        self.param_groups = [dict(
            lr=lr, in_momentum=in_momentum, out_momentum=out_momentum,
            dampening=dampening, weight_decay=weight_decay, nesterov=nesterov,
            params=[arena.flat])]
        self._in_buf = None
        self._out_buf = None
        self._in_init = False
        self._out_init = False
        # fused corrections (flat buffers or None)
        self._prox_mu = 0.0
        self._server = None
        self._ctrl_server = None
        self._ctrl_client = None
        self._delta = None

    # ---- virtual-client state swap -----------------------------------------
    def bind_state(self, in_buf=None, out_buf=None, in_init=True,
                   out_init=True):
        """Point the momentum buffers at externally owned per-client slices
        (virtual-client packing keeps [C, N] momentum arenas resident in
        HBM and swaps views per client)."""
        self._in_buf = in_buf
        self._out_buf = out_buf
        self._in_init = in_init
        self._out_init = out_init

    # ---- correction plumbing ------------------------------------------------
    def set_correction(self, prox_mu=0.0, server=None, ctrl_server=None,
                       ctrl_client=None, delta=None):
        self._prox_mu = prox_mu
        self._server = server
        self._ctrl_server = ctrl_server
        self._ctrl_client = ctrl_client
        self._delta = delta

    def clear_correction(self):
        self.set_correction()

    # ---- optimizer API ------------------------------------------------------
    def zero_grad(self, set_to_none=False):
        self.arena.zero_grad()

    def step(self, closure=None, apply_lr=True, scale=1.0,
             apply_in_momentum=True, apply_out_momentum=False, grad=None):
        """One fused step.  ``grad``: optional flat tensor to use instead of
        the arena gradient (the aggregation sync step passes the aggregated
        diff here instead of clobbering ``p.grad`` like the reference does)."""
        loss = closure() if closure is not None else None
        g = self.param_groups[0]
        use_in = g['in_momentum'] != 0 and apply_in_momentum
        use_out = g['out_momentum'] != 0 and apply_out_momentum
        if use_in and self._in_buf is None:
            self._in_buf = self.arena.new_buffer()
        if use_out and self._out_buf is None:
            self._out_buf = self.arena.new_buffer()
        # per-algorithm corrections apply to LOCAL steps only — the sync
        # step applies the raw aggregate (reference corrects p.grad inside
        # the local loop, `trainings/federated/main.py:116-129`, never at
        # aggregation time).
        local = apply_lr
        ops.fused_sgd_step(
            self.arena.flat,
            grad if grad is not None else self.arena.grad,
            lr=g['lr'], scale=scale, weight_decay=g['weight_decay'],
            in_momentum=g['in_momentum'], out_momentum=g['out_momentum'],
            dampening=g['dampening'], nesterov=g['nesterov'],
            apply_lr=apply_lr, apply_in_momentum=use_in,
            apply_out_momentum=use_out,
            in_buf=self._in_buf, out_buf=self._out_buf,
            first_in=use_in and not self._in_init,
            first_out=use_out and not self._out_init,
            prox_mu=self._prox_mu if local else 0.0,
            server=self._server if local else None,
            ctrl_server=self._ctrl_server if local else None,
            ctrl_client=self._ctrl_client if local else None,
            delta=self._delta if local else None,
            wd_numel=self.arena.wd_numel,
            half_param=getattr(self.arena, 'half_flat', None))
        if use_in:
            self._in_init = True
        if use_out:
            self._out_init = True
        return loss

    # ---- state --------------------------------------------------------------
    def state_dict(self):
        g = self.param_groups[0]
        return {
            'group': {k: v for k, v in g.items() if k != 'params'},
            'in_buf': self._in_buf, 'out_buf': self._out_buf,
            'in_init': self._in_init, 'out_init': self._out_init,
        }

    def load_state_dict(self, sd):
        self.param_groups[0].update(sd['group'])
        self._in_buf = None if sd['in_buf'] is None else \
            sd['in_buf'].to(self.arena.flat.device)
        self._out_buf = None if sd['out_buf'] is None else \
            sd['out_buf'].to(self.arena.flat.device)
        self._in_init = sd['in_init']
        self._out_init = sd['out_init']
