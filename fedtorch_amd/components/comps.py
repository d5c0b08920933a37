# -*- coding: utf-8 -*-
"""Component factory (parity with reference `components/comps.py:10-33`).

Order matters exactly like the reference: the scheduler factory writes
``args.learning_rate`` (`scheduler.py:51-55`) which the optimizer factory
consumes.  The MI355X twist: the model is moved to the GPU FIRST and then
wrapped in a flat :class:`~fedtorch_amd.parallel.arena.Arena`; the optimizer
operates on the arena.
"""
import torch

from fedtorch_amd.components.model import define_model
from fedtorch_amd.components.scheduler import define_lr_scheduler
from fedtorch_amd.components.optimizer import define_optimizer
from fedtorch_amd.components.criterion import define_criterion
from fedtorch_amd.components.metrics import define_metrics
from fedtorch_amd.parallel.arena import Arena
from fedtorch_amd.logs.checkpoint import maybe_resume_from_checkpoint


def create_components(args):
    """Returns (model, arena, criterion, scheduler, optimizer, metrics)."""
    model = define_model(args)
    if args.graph.on_cuda and torch.cuda.is_available():
        model = model.cuda()
        cl = getattr(args, 'channels_last', False)
        if cl:
            model = model.to(memory_format=torch.channels_last)
        if getattr(args, 'fused_bn', True) and \
                getattr(args, 'hip_kernels', True):
            from fedtorch_amd.ops.batchnorm import convert_to_fused_bn
            import fedtorch_amd.ops as _ops
            if _ops.hip_available():
                convert_to_fused_bn(model)
                if cl:
                    # custom NHWC stem kernels (MIOpen's NHWC bf16 path
                    # falls to naive_conv for 3-channel stems)
                    from fedtorch_amd.ops.stemconv import convert_stem
                    convert_stem(model)
    arena = Arena(model)
    criterion = define_criterion(args)
    if args.graph.on_cuda and torch.cuda.is_available():
        criterion = criterion.cuda()
    # scheduler BEFORE optimizer (writes args.learning_rate).
    scheduler = define_lr_scheduler(args)
    optimizer = define_optimizer(args, arena)
    metrics = define_metrics(args, model)
    maybe_resume_from_checkpoint(args, model, optimizer)
    return model, arena, criterion, scheduler, optimizer, metrics
