# -*- coding: utf-8 -*-
"""Local-SGD sync scheme + aggregation
(parity with reference `comms/algorithms/distributed.py`).

`aggregate_gradients` is the arena version of `distributed.py:108-142`:
ONE fused diff+restore kernel, ONE RCCL all-reduce over the arena, ONE fused
apply step — instead of P per-parameter loops/messages.
"""
import numpy as np
import torch
import torch.distributed as dist

from fedtorch_amd import ops


def configure_sync_scheme(args):
    args.local_steps = define_sync_freq(
        num_epochs=args.num_epochs,
        local_step=args.local_step,
        local_step_warmup_type=args.local_step_warmup_type,
        local_step_warmup_period=args.local_step_warmup_period,
        turn_on_local_step_from=args.turn_on_local_step_from,
        turn_off_local_step_from=args.turn_off_local_step_from,
        warmup_per_intervals=args.local_step_warmup_per_interval,
        lr_change_epochs=args.lr_change_epochs)


def define_sync_freq(num_epochs, local_step, local_step_warmup_type,
                     local_step_warmup_period, turn_on_local_step_from,
                     turn_off_local_step_from, warmup_per_intervals,
                     lr_change_epochs):
    """Per-epoch list of local steps, incl. warmup ramps and on/off windows
    (reference `distributed.py:17-106` — semantics preserved exactly)."""
    num_epochs = num_epochs + 2
    if local_step_warmup_period is None:
        local_step_warmup_period = local_step

    if local_step_warmup_type is None:
        tmp_steps = [local_step] * local_step_warmup_period
    elif 'exp' in local_step_warmup_type:
        log_local_step = int(np.log2(local_step_warmup_period))
        tmp_steps = [2 ** int(ind * log_local_step / local_step_warmup_period)
                     for ind in range(1, 1 + local_step_warmup_period)]
    elif 'linear' in local_step_warmup_type:
        tmp_steps = [max(1, int(ind * local_step / local_step_warmup_period))
                     for ind in range(1, 1 + local_step_warmup_period)]
    elif 'constant' in local_step_warmup_type:
        tmp_steps = [1] * local_step_warmup_period
    else:
        raise NotImplementedError(local_step_warmup_type)
    if len(tmp_steps) > num_epochs:
        tmp_steps = tmp_steps[:num_epochs]

    if lr_change_epochs is not None:
        changes = [int(x) for x in lr_change_epochs.split(',')]
        changes = [0] + changes + [num_epochs]
        fromto = list(zip(changes[:-1], changes[1:]))

    if not warmup_per_intervals:
        steps = []
        if lr_change_epochs is None:
            steps = tmp_steps + [local_step] * (num_epochs - len(tmp_steps))
        else:
            if turn_on_local_step_from is None and \
                    turn_off_local_step_from is None:
                return tmp_steps + [local_step] * (num_epochs - len(tmp_steps))
            for from_ind, to_ind in fromto:
                if turn_on_local_step_from is None and \
                        turn_off_local_step_from is not None:
                    if from_ind >= turn_off_local_step_from:
                        steps += [1] * (to_ind - from_ind)
                    else:
                        steps += [local_step] * (to_ind - from_ind)
                elif turn_on_local_step_from is not None and \
                        turn_off_local_step_from is None:
                    if from_ind >= turn_on_local_step_from:
                        steps += [local_step] * (to_ind - from_ind)
                    else:
                        steps += [1] * (to_ind - from_ind)
                else:
                    raise NotImplementedError(
                        'both turn_on and turn_off set')
    else:
        steps = []
        for from_ind, to_ind in fromto:
            t = [local_step] * (to_ind - from_ind - len(tmp_steps))
            steps += tmp_steps + t
    return steps


def aggregate_gradients(args, comm, arena, old_flat, optimizer, agg_buf):
    """Periodic model-diff all-reduce for local SGD (reference
    `distributed.py:108-142`)."""
    # diff = old - current (weight 1), restore current = old; one kernel.
    ops.weighted_diff_restore(old_flat, arena.flat, agg_buf, 1.0)
    comm.all_reduce(agg_buf)
    if args.avg_model:
        agg_buf.div_(float(args.graph.n_nodes))
    optimizer.step(apply_lr=False, scale=args.lr_scale_at_sync,
                   apply_in_momentum=False,
                   apply_out_momentum=args.out_momentum, grad=agg_buf)
    old_flat.copy_(arena.flat)
    return old_flat


def global_average(sum_, count, group=None):
    """2-element all-reduce metric averaging (reference
    `distributed.py:148-161`)."""
    array = torch.tensor([float(sum_), float(count)], dtype=torch.float32)
    if dist.is_available() and dist.is_initialized():
        if group is None:
            dist.all_reduce(array, op=dist.ReduceOp.SUM)
        else:
            dist.all_reduce(array, op=dist.ReduceOp.SUM, group=group)
    all_sum, all_count = array
    if all_count == 0:
        return torch.tensor(0.0)
    return all_sum / all_count


def elementwise_min(tensor, group=None):
    if dist.is_available() and dist.is_initialized():
        dist.all_reduce(tensor, op=dist.ReduceOp.MIN)
    return tensor
