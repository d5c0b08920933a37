# -*- coding: utf-8 -*-
"""Centered validation (parity with reference `comms/utils/eval_centered.py`)."""
import numpy as np
import torch

from fedtorch_amd.components.dataset import _load_data_batch
from fedtorch_amd.trainings.eval import inference, inference_personal
from fedtorch_amd.logs.logging import log, update_performancec_tracker


def do_validate_centered(args, model, criterion, metrics, optimizer,
                         val_loader, val_tracker, personal=False, val=True,
                         model_personal=None, alpha=0.0, local=False):
    if 'robust' in args.arch:
        tmp_noise = torch.clone(model.noise.data)
        model.noise.data.zero_()
        for _input, _target in val_loader:
            _input, _target = _load_data_batch(args, _input, _target)
            loss, _ = inference(model, criterion, metrics, _input, _target)
            grad = torch.autograd.grad(loss, model.noise)[0]
            model.noise.data.add_(grad, alpha=0.01)
            nrm = torch.norm(model.noise.data)
            if nrm > 1:
                model.noise.data.div_(nrm)
    model.eval()
    if personal:
        if model_personal is None:
            raise ValueError('model_personal required')
        model_personal.eval()
    for _input, _target in val_loader:
        _input, _target = _load_data_batch(args, _input, _target)
        if _input.size(0) == 1:
            break
        with torch.no_grad():
            if personal:
                loss, performance = inference_personal(
                    model_personal, model, alpha, criterion, metrics,
                    _input, _target)
            else:
                loss, performance = inference(
                    model, criterion, metrics, _input, _target,
                    rnn=args.arch == 'rnn')
            update_performancec_tracker(val_tracker, loss, performance,
                                        _input.size(0))
    model.train()
    if personal:
        model_personal.train()
    if 'robust' in args.arch:
        model.noise.data.copy_(tmp_noise)


def log_validation_centered(args, val_tracker, personal=False, val=True,
                            local=False):
    performance = [val_tracker[x].avg for x in ['top1', 'top5', 'losses']]
    p0 = 'Personal' if personal or local else 'Global'
    p1 = 'validation' if val else 'train'
    log('{} performance for {} at batch: {}. Epoch: {}. Process: {}. '
        'Prec@1: {:.3f} Prec@5: {:.3f} Loss: {:.3f} Comm: {}'.format(
            p0, p1, args.local_index, args.epoch, args.graph.rank,
            performance[0], performance[1], performance[2],
            args.rounds_comm), debug=args.debug)


def log_validation_per_client_centered(args, Clients, online_clients,
                                       val=True, local=False):
    acc = []
    for oc in online_clients:
        if local:
            t = Clients[oc].local_personal_val_tracker if val else \
                Clients[oc].local_val_tracker
        else:
            t = Clients[oc].global_personal_val_tracker if val else \
                Clients[oc].global_val_tracker
        acc.append(t['top1'].avg)
    log('{} per client stat for {} at batch: {}. Epoch: {}. Process: {}. '
        'Worst: {:.3f} Best: {:.3f} Var: {:.3f} Comm: {}'.format(
            'Personal' if local else 'Global',
            'validation' if val else 'train', args.local_index, args.epoch,
            args.graph.rank, np.min(acc), np.max(acc), np.std(acc),
            args.rounds_comm), debug=args.debug)


def log_test_centered(args, val_tracker):
    performance = [val_tracker[x].avg for x in ['top1', 'top5', 'losses']]
    log('Test at batch: {}. Epoch: {}. Process: {}. Prec@1: {:.3f} '
        'Prec@5: {:.3f} Loss: {:.3f} Comm: {}'.format(
            args.local_index, args.epoch, args.graph.rank, performance[0],
            performance[1], performance[2], args.rounds_comm),
        debug=args.debug)
