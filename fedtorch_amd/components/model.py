# -*- coding: utf-8 -*-
"""Model factory (parity with reference `components/model.py`)."""
import torch
import torch.distributed as dist

import fedtorch_amd.components.models as models
from fedtorch_amd.logs.logging import log


def define_model(args):
    # order matters: 'wideresnet' contains 'resnet'
    if 'wideresnet' in args.arch:
        model = models.wideresnet(args)
    elif 'densenet' in args.arch:
        model = models.densenet(args)
    elif 'resnet' in args.arch:
        model = models.resnet(args)
    elif args.arch in ('mlp', 'robust_mlp', 'cnn', 'rnn',
                       'logistic_regression', 'robust_logistic_regression',
                       'least_square', 'robust_least_square'):
        model = getattr(models, args.arch)(args)
    else:
        raise NotImplementedError('unknown arch: %s' % args.arch)
    get_model_stat(args, model)
    return model


def get_model_stat(args, model):
    n_params = sum(p.numel() for p in model.parameters())
    log('=> creating model {} with {:.2f}K parameters'.format(
        args.arch, n_params / 1e3), debug=args.debug)
    return n_params


def consistent_model(args, model):
    """Make rank 0's init authoritative on every rank.

    The reference zeroes non-rank-0 params then all-reduduces SUM per param
    (`components/model.py:33-43`, xP messages); here it is ONE broadcast of
    the flat arena if the model is arena-backed, else per-tensor broadcast.
    """
    if not (dist.is_available() and dist.is_initialized()):
        return model
    log('consistent model across ranks (broadcast from 0)', debug=args.debug)
    for p in model.parameters():
        dist.broadcast(p.data, src=0)
    return model


def consistent_arena(arena):
    """One flat broadcast: every rank takes rank 0's arena."""
    if dist.is_available() and dist.is_initialized():
        dist.broadcast(arena.flat, src=0)
    return arena
