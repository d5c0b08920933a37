# -*- coding: utf-8 -*-
"""Accuracy metrics (parity with reference `components/metrics.py`)."""
import torch

from fedtorch_amd.logs.meter import AverageMeter


def define_metrics(args, model):
    if 'least_square' in args.arch:
        return ()
    if args.arch == 'rnn':
        return (1,)
    if getattr(model, 'num_classes', 2) >= 5:
        return (1, 5)
    return (1,)


class TopKAccuracy(object):
    def __init__(self, topk=1):
        self.topk = topk
        self.reset()

    def __call__(self, output, target):
        batch_size = target.size(0)
        _, pred = output.topk(self.topk, 1, True, True)
        pred = pred.t()
        correct = pred.eq(target.view(1, -1).expand_as(pred))
        correct_k = correct[:self.topk].reshape(-1).float().sum(0, keepdim=True)
        return correct_k.mul_(100.0 / batch_size)

    def reset(self):
        self.top = AverageMeter()

    def update(self, prec, size):
        self.top.update(prec, size)

    def average(self):
        from fedtorch_amd.aggregation.distributed import global_average
        return global_average(self.top.sum, self.top.count)

    @property
    def name(self):
        return 'Prec@{}'.format(self.topk)


def accuracy(output, target, topk=(1,), rnn=False):
    """precision@k (reference `metrics.py:50-73`)."""
    res = []
    if not rnn:
        if len(topk) > 0:
            maxk = max(topk)
            batch_size = target.size(0)
            _, pred = output.topk(maxk, 1, True, True)
            pred = pred.t()
            correct = pred.eq(target.view(1, -1).expand_as(pred))
            for k in topk:
                correct_k = correct[:k].reshape(-1).float().sum(0, keepdim=True)
                res.append(correct_k.mul_(100.0 / batch_size).item())
        else:
            res += [0]
    else:
        pred = output.argmax(dim=1, keepdim=True)
        correct = pred.eq(target.view_as(pred)).float().mean()
        res.append(correct.mul_(100.0).item())
    return res


def accuracy_per_class(output, target, classes):
    """per-class precision (reference `metrics.py:77-91`)."""
    acc = torch.zeros_like(classes).float()
    count = torch.zeros_like(classes).float()
    _, pred = torch.max(output, 1)
    target = torch.squeeze(target)
    correct = pred.eq(target)
    for i, c in enumerate(classes):
        c_inds = target == c
        count[i] = c_inds.float().sum()
        acc[i] = 0.0 if count[i] == 0 else \
            (c_inds & correct).float().sum().mul_(100.0 / count[i])
    return acc, count
