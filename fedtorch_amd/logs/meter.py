# -*- coding: utf-8 -*-
"""Meters (parity with reference `fedtorch/logs/meter.py`).

Cross-rank averaging is done by ONE fused all-reduce over a small buffer
instead of a 2-float all-reduce per meter (reference
`comms/algorithms/distributed.py:148-161` called per meter): see
`evaluate_global_performance_fused`.
"""
import torch


class AverageMeter(object):
    """Computes and stores the average and current value."""

    def __init__(self):
        self.reset()

    def reset(self):
        self.val = 0
        self.avg = 0
        self.sum = 0
        self.count = 0

    def update(self, val, n=1):
        self.val = val
        self.sum += val * n
        self.count += n
        self.avg = self.sum / self.count if self.count != 0 else 0


def define_trackers(names):
    return dict((name, AverageMeter()) for name in names)


def define_local_training_tracker():
    return define_trackers([
        'computing_time', 'global_time', 'data_time',
        'sync_time', 'load_time', 'losses', 'top1', 'top5', 'learning_rate'])


def define_val_tracker():
    return define_trackers(['losses', 'top1', 'top5'])


def define_per_class_acc_tracker(classes):
    return define_trackers(list(classes))


def evaluate_local_performance(meter):
    return meter.sum / meter.count if meter.count else 0


def evaluate_gloabl_performance(meter, group=None):
    """Reference-spelled name kept for parity (`meter.py:23-24`)."""
    from fedtorch_amd.aggregation.distributed import global_average
    return global_average(meter.sum, meter.count, group)


def evaluate_global_performance_fused(meters, group=None):
    """All meters averaged with ONE all-reduce of a 2xM float tensor."""
    import torch.distributed as dist
    buf = torch.tensor([[m.sum for m in meters], [float(m.count) for m in meters]],
                       dtype=torch.float64)
    if dist.is_available() and dist.is_initialized():
        dist.all_reduce(buf, op=dist.ReduceOp.SUM, group=group)
    sums, counts = buf[0], buf[1]
    return [float(s / c) if c != 0 else 0.0 for s, c in zip(sums, counts)]
