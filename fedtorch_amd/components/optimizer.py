# -*- coding: utf-8 -*-
"""Optimizer factory (parity with reference `components/optimizer.py`).

The reference builds one param-group per tensor with wd=0 for 'bn' params
(`optimizer.py:8-16`); the arena encodes that as a wd prefix instead (see
`fedtorch_amd/parallel/arena.py`), so the fused step stays one kernel.
Out-momentum default is ``1 - 1/n_nodes`` (`optimizer.py:23-25`).
"""
from fedtorch_amd.components.optim.sgd import FusedSGD
from fedtorch_amd.components.optim.adam import FusedAdamW


def define_optimizer(args, arena):
    if args.optimizer == 'sgd':
        return FusedSGD(
            arena, lr=args.learning_rate,
            in_momentum=args.in_momentum_factor,
            out_momentum=(args.out_momentum_factor
                          if args.out_momentum_factor is not None
                          else 1.0 - 1.0 / args.graph.n_nodes),
            weight_decay=args.weight_decay,
            nesterov=args.use_nesterov)
    return FusedAdamW(arena, lr=args.learning_rate,
                      weight_decay=args.weight_decay,
                      correct_wd=args.correct_wd)
