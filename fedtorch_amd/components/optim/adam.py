# -*- coding: utf-8 -*-
"""AdamW over a flat arena (parity with reference
`components/optimizers/adam.py:48-104`, including its dual-use convention:
``apply_lr=False`` applies ``p -= scale * grad`` BEFORE the moment updates,
which is how the reference uses it at sync).

Documented divergence: the reference (`adam.py:97-99`) applies the
moment-based ``addcdiv`` step UNCONDITIONALLY, i.e. even on the sync call
(`apply_lr=False`) it takes a second, lr-sized moment step on top of
``p -= scale*grad``.  Here the moment step is gated on ``apply_lr`` — the
sync call only applies the scaled aggregate (and still refreshes the
moments with it), which is the behavior every other optimizer in both
codebases has at sync.  This changes adam/fedadam sync trajectories
relative to the reference by design (like the other deliberate bug fixes
listed in `aggregation/federated.py:14-25`)."""
import math

import torch




class FusedAdamW(object):
    def __init__(self, arena, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=0.0, correct_wd=False):
        self.arena = arena
        self.param_groups = [dict(lr=lr, betas=betas, eps=eps,
                                  weight_decay=weight_decay,
                                  correct_wd=correct_wd,
                                  params=[arena.flat])]
        self._exp_avg = None
        self._exp_avg_sq = None
        self._step = 0

    def zero_grad(self, set_to_none=False):
        self.arena.zero_grad()

    def step(self, closure=None, apply_lr=True, scale=1.0,
             apply_in_momentum=True, apply_out_momentum=False, grad=None):
        loss = closure() if closure is not None else None
        g = self.param_groups[0]
        p = self.arena.flat
        gr = grad if grad is not None else self.arena.grad
        if self._exp_avg is None:
            self._exp_avg = self.arena.new_buffer()
            self._exp_avg_sq = self.arena.new_buffer()
        if not apply_lr:
            # reference `adam.py:71-72`: sync applies the scaled aggregate
            # directly, then still updates the moments with it.
            p.add_(gr, alpha=-scale)
        beta1, beta2 = g['betas']
        nw = self.arena.wd_numel
        if g['weight_decay'] != 0 and not g['correct_wd']:
            gr = gr.clone()
            gr[:nw].add_(p[:nw], alpha=g['weight_decay'])
        self._exp_avg.mul_(beta1).add_(gr, alpha=1 - beta1)
        self._exp_avg_sq.mul_(beta2).addcmul_(gr, gr, value=1 - beta2)
        denom = self._exp_avg_sq.sqrt().add_(g['eps'])
        self._step += 1
        bc1 = 1 - beta1 ** self._step
        bc2 = 1 - beta2 ** self._step
        step_size = g['lr'] * math.sqrt(bc2) / bc1
        if not g['correct_wd']:
            if apply_lr:
                p.addcdiv_(self._exp_avg, denom, value=-step_size)
        else:
            upd = torch.zeros_like(p)
            upd[:nw] = p[:nw] * (-step_size * g['weight_decay'])
            upd.addcdiv_(self._exp_avg, denom, value=1)
            if apply_lr:
                p.add_(upd)
        return loss

    def state_dict(self):
        g = self.param_groups[0]
        return {'group': {k: v for k, v in g.items() if k != 'params'},
                'exp_avg': self._exp_avg, 'exp_avg_sq': self._exp_avg_sq,
                'step': self._step}

    def load_state_dict(self, sd):
        self.param_groups[0].update(sd['group'])
        dev = self.arena.flat.device
        self._exp_avg = None if sd['exp_avg'] is None else sd['exp_avg'].to(dev)
        self._exp_avg_sq = None if sd['exp_avg_sq'] is None else \
            sd['exp_avg_sq'].to(dev)
        self._step = sd['step']
