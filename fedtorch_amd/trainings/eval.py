# -*- coding: utf-8 -*-
"""Validation paths (parity with reference `comms/utils/eval.py`)."""
import torch

from fedtorch_amd.components.metrics import accuracy, accuracy_per_class
from fedtorch_amd.components.dataset import _load_data_batch
from fedtorch_amd.logs.checkpoint import save_to_checkpoint
from fedtorch_amd.logs.logging import (
    logging_display_val, logging_display_test_summary,
    update_performancec_tracker)
from fedtorch_amd.logs.meter import (
    define_val_tracker, evaluate_gloabl_performance,
    evaluate_local_performance)


def inference(model, criterion, metrics, _input, _target, classes=None,
              rnn=False):
    """forward + loss + accuracy (reference `eval.py:17-29`)."""
    output = model(_input)
    loss = criterion(output, _target)
    performance = accuracy(output.data, _target, topk=metrics, rnn=rnn)
    if classes is not None:
        acc_pc, count_pc = accuracy_per_class(output.data, _target, classes)
        return loss, performance, (acc_pc, count_pc)
    return loss, performance


def inference_personal(model1, model2, alpha, criterion, metrics, _input,
                       _target):
    """alpha-blend of two models' logits (reference `eval.py:31-39`)."""
    output = alpha * model1(_input) + (1 - alpha) * model2(_input)
    loss = criterion(output, _target)
    performance = accuracy(output.data, _target, topk=metrics)
    return loss, performance


def do_validate(args, model, optimizer, criterion, metrics, data_loader,
                group, data_mode='validation', personal=False,
                model_personal=None, alpha=0.0, local=False, skip=False):
    """Evaluate; on test mode rank 0 checkpoints on new best
    (reference `eval.py:41-150`).

    ``skip=True``: participate in the metric collectives with an empty
    tracker but do no local evaluation — used by OFFLINE ranks, which in the
    reference sit outside the online group (`trainings/federated/main.py`)
    but here join the world collective with zero weight.
    """
    model_mode = 'personal' if personal or local else 'global'
    tracker = define_val_tracker()

    if skip:
        for x in ('top1', 'top5', 'losses'):
            tracker[x].sum, tracker[x].count = 0.0, 0
        performance = [float(evaluate_gloabl_performance(tracker[x], group))
                       for x in ['top1', 'top5', 'losses']]
        return performance

    if 'robust' in args.arch:
        # adversarial noise ascent (reference `eval.py:59-68`)
        tmp_noise = torch.clone(model.noise.data)
        for _input, _target in data_loader:
            _input, _target = _load_data_batch(args, _input, _target)
            loss, _ = inference(model, criterion, metrics, _input, _target)
            grad = torch.autograd.grad(loss, model.noise)[0]
            model.noise.data.add_(grad, alpha=0.01)
            nrm = torch.norm(model.noise.data)
            if nrm > 1:
                model.noise.data /= nrm

    model.eval()
    if personal:
        if model_personal is None:
            raise ValueError('model_personal required for personalized '
                             'validation (APFL)')
        model_personal.eval()
    for _input, _target in data_loader:
        _input, _target = _load_data_batch(args, _input, _target)
        if _input.size(0) == 1:
            break  # BatchNorm issue (reference `eval.py:89-91`)
        # bf16 twin weights (arena.enable_bf16_compute, hipGraph paths)
        # need autocast in eval too: fp32 inputs vs bf16 conv weights
        from fedtorch_amd.trainings.graphstep import amp as _amp
        with torch.no_grad(), _amp(args):
            if personal:
                loss, performance = inference_personal(
                    model_personal, model, alpha, criterion, metrics,
                    _input, _target)
            else:
                loss, performance = inference(
                    model, criterion, metrics, _input, _target,
                    rnn=args.arch == 'rnn')
            tracker = update_performancec_tracker(
                tracker, loss, performance, _input.size(0))
    model.train()
    if len(metrics) == 1:
        tracker['top5'].count = 1.0
        tracker['top5'].sum = 0.0
        tracker['top5'].avg = 0.0
    if data_mode == 'test' and model_mode == 'global':
        performance = [evaluate_local_performance(tracker[x])
                       for x in ['top1', 'top5', 'losses']]
    else:
        performance = [float(evaluate_gloabl_performance(tracker[x], group))
                       for x in ['top1', 'top5', 'losses']]

    logging_display_val(args, performance, mode=data_mode,
                        personal=model_mode == 'personal')

    if data_mode == 'test' and not personal:
        args.cur_prec1 = performance[0]
        is_best = args.cur_prec1 > args.best_prec1
        if is_best:
            args.best_prec1 = performance[0]
            args.best_epoch += [args.epoch_]
        logging_display_test_summary(args, debug=args.debug)
        if args.graph.rank == 0 and args.debug:
            state = {
                'arguments': _checkpointable_args(args),
                'current_epoch': args.epoch,
                'local_index': args.local_index,
                'global_index': args.global_index,
                'arch': args.arch,
                'state_dict': model.state_dict(),
                'optimizer': optimizer.state_dict(),
                'best_prec1': args.best_prec1,
            }
            save_to_checkpoint(state, is_best, dirname=args.checkpoint_root,
                               filename='checkpoint.pth.tar',
                               save_all=args.save_all_models)

    if 'robust' in args.arch:
        # copy_, not assignment: noise is a view into the arena.
        model.noise.data.copy_(tmp_noise)
    return performance


def _checkpointable_args(args):
    """args minus unpicklable run-time objects (graph keeps its repr)."""
    import copy
    a = copy.copy(args)
    return a
