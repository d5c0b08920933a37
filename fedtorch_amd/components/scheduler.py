# -*- coding: utf-8 -*-
"""Learning-rate scheduling (parity with reference `components/scheduler.py`
and `components/optimizers/learning.py`).

Epoch indices are FRACTIONAL (driven by `state.epoch_` =
local_index / num_batches_per_epoch) and the scheduler is re-evaluated every
local step (reference `scheduler.py:9-29`).  Schemes: `strict`,
`custom_one_cycle`, `custom_multistep`, `custom_convex_decay`
(`learning.py:13-25`), with per-interval linear / poly / convex
(gamma/(mu(t+a))) scales (`learning.py:211-228`).
"""


def define_scheduler(args):
    return define_lr_scheduler(args)


def adjust_learning_rate(args, optimizer, lr_scheduler, lr_external=None):
    """Write the scheduled lr into the optimizer (reference `scheduler.py:9-29`)."""
    if lr_external is None:
        lr = lr_scheduler(args.epoch_)
        if lr is None:
            lr = args.old_learning_rate
        if args.old_learning_rate != lr:
            args.old_learning_rate = lr
            for param_group in optimizer.param_groups:
                param_group['lr'] = lr
    else:
        for param_group in optimizer.param_groups:
            param_group['lr'] = lr_external
        lr = lr_external
    return lr


def define_lr_scheduler(args):
    """Computes args.learning_rate (with linear/sqrt scale-up) and returns the
    schedule function (reference `scheduler.py:32-62`).  MUST run before the
    optimizer factory — it writes `args.learning_rate`."""
    args.learning_rate_per_samples = args.lr / args.batch_size
    args.init_warmup_lr = args.lr

    if args.lr_scaleup:
        if args.lr_scaleup_type == 'linear':
            _lr = args.learning_rate_per_samples * args.batch_size
            _scale = args.graph.n_nodes
        elif args.lr_scaleup_type == 'sqrt':
            _lr = args.lr
            _scale = (1. * args.graph.n_nodes * args.batch_size /
                      args.base_batch_size) ** 0.5
        else:
            raise NotImplementedError(args.lr_scaleup_type)
        args.learning_rate = _lr * _scale
    else:
        args.learning_rate = args.learning_rate_per_samples * args.batch_size

    args.old_learning_rate = args.learning_rate
    return get_lr_scheduler(args)


# --------------------------------------------------------------------------
# schedule construction (reference `learning.py`)
# --------------------------------------------------------------------------

def get_lr_scheduler(args):
    epoch_fields, lr_fields, scale_indicators = get_scheduling_setup(args)
    schedulers = [
        _build_lr_scheduler(args, ef, lf, ind)
        for ef, lf, ind in zip(epoch_fields, lr_fields, scale_indicators)]

    def f(epoch_index):
        for (lo, hi), sched in zip(epoch_fields, schedulers):
            if lo <= epoch_index < hi:
                return sched(epoch_index)
        return None
    return f


def get_scheduling_setup(args):
    scheme = args.lr_schedule_scheme
    if scheme == 'strict':
        args.lr_change_epochs = '0,{},{}'.format(
            args.lr_change_epochs, args.num_epochs)
        return _parse_setup(args)
    if scheme == 'custom_one_cycle':
        args.lr_fields = '{low},{high}/{high},{low}/{low},{extra_low}'.format(
            low=args.lr_onecycle_low, high=args.lr_onecycle_high,
            extra_low=args.lr_onecycle_extra_low)
        args.lr_change_epochs = '0,{},{},{}'.format(
            args.lr_onecycle_num_epoch // 2, args.lr_onecycle_num_epoch,
            args.num_epochs)
        args.lr_scale_indicators = '0,0,0'
        return _parse_setup(args)
    if scheme == 'custom_multistep':
        args.lr_fields = _build_multistep_lr_fields(
            args.lr_change_epochs, args.lr_warmup, args.learning_rate,
            args.init_warmup_lr, args.lr_decay)
        args.lr_change_epochs, n_intervals = _build_multistep_lr_change_epochs(
            args.lr_change_epochs, args.lr_warmup, args.lr_warmup_epochs,
            args.num_epochs)
        args.lr_scale_indicators = ','.join(['0'] * n_intervals)
        return _parse_setup(args)
    if scheme == 'custom_convex_decay':
        args.lr_fields = '{},{}'.format(args.learning_rate, 0)
        args.lr_change_epochs = '0,{}'.format(args.num_epochs)
        args.lr_scale_indicators = '2'
        return _parse_setup(args)
    if scheme is None:
        # constant lr — reference requires a scheme, but a constant default
        # makes the engine usable without schedule flags.
        args.lr_fields = '{lr},{lr}'.format(lr=args.learning_rate)
        args.lr_change_epochs = '0,{}'.format(max(args.num_epochs, 1))
        args.lr_scale_indicators = '0'
        return _parse_setup(args)
    raise NotImplementedError(scheme)


def _parse_setup(args):
    lr_fields = [list(map(float, f.split(',')))
                 for f in args.lr_fields.split('/')]
    indicators = [{'0': 'linear', '1': 'poly', '2': 'convex'}[i]
                  for i in args.lr_scale_indicators.split(',')]
    changes = [int(x) for x in args.lr_change_epochs.split(',')]
    epoch_fields = list(zip(changes[:-1], changes[1:]))
    return epoch_fields, lr_fields, indicators


def _build_multistep_lr_fields(lr_change_epochs, lr_warmup, learning_rate,
                               init_warmup_lr, lr_decay):
    if lr_change_epochs is not None:
        _lr_fields = [learning_rate * ((1. / lr_decay) ** l)
                      for l in range(len(lr_change_epochs.split(',')) + 1)]
    else:
        _lr_fields = [learning_rate]
    fields = '/'.join(['{lr},{lr}'.format(lr=lr) for lr in _lr_fields])
    if lr_warmup:
        return '{},{}/'.format(init_warmup_lr, learning_rate) + fields
    return fields


def _build_multistep_lr_change_epochs(lr_change_epochs, lr_warmup,
                                      lr_warmup_epochs, num_epochs):
    if lr_change_epochs is not None:
        lr_change_epochs = [0] + lr_change_epochs.split(',') + [num_epochs]
    else:
        lr_change_epochs = [0, num_epochs]
    if lr_warmup:
        lr_change_epochs = [0, lr_warmup_epochs] + lr_change_epochs[1:]
    return (','.join([str(x) for x in lr_change_epochs]),
            len(lr_change_epochs) - 1)


def _build_lr_scheduler(args, epoch_field, lr_field, scale_indicator):
    lr_left, lr_right = lr_field
    epoch_left, epoch_right = epoch_field
    n_steps = epoch_right - epoch_left
    if scale_indicator == 'linear':
        def f(index):
            step = (lr_right - lr_left) / n_steps
            return (index - epoch_left) * step + lr_left
        return f
    if scale_indicator == 'poly':
        def f(index):
            return lr_left * ((1 - (index - epoch_left) / n_steps) ** 2)
        return f
    if scale_indicator == 'convex':
        assert args.lr_gamma is not None
        assert args.lr_mu is not None
        assert args.lr_alpha is not None
        gamma, mu, alpha = args.lr_gamma, args.lr_mu, args.lr_alpha

        def f(index):
            return gamma / (mu * (alpha + index))
        return f
    raise NotImplementedError(scale_indicator)
