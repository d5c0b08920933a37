# -*- coding: utf-8 -*-
"""APFL — Adaptive Personalized Federated Learning (arXiv:2003.13461);
parity with reference `comms/trainings/federated/apfl.py:30-180`.

Per local step TWO forward/backward passes: the global model steps normally,
then the personal model steps on the alpha-blended loss; optional adaptive
alpha once per round (fused arena dot-product, `ops.alpha_grad`).  Sync
aggregates only the global model via FedAvg.
"""
import time

import numpy as np
import torch

from fedtorch_amd import ops
from fedtorch_amd.components.scheduler import adjust_learning_rate
from fedtorch_amd.components.dataset import load_data_batch
from fedtorch_amd.trainings.flow import get_current_epoch, is_sync_fed
from fedtorch_amd.trainings.eval import (inference, inference_personal,
                                         do_validate)
from fedtorch_amd.trainings.federated import amp
from fedtorch_amd.aggregation.federated import (fedavg_aggregation,
                                                distribute_model_server)
from fedtorch_amd.aggregation.distributed import global_average
from fedtorch_amd.logs.logging import (
    log, logging_sync_time, logging_load_time, logging_globally)
from fedtorch_amd.logs.meter import define_local_training_tracker


def apfl_alpha_update(client, lr):
    """alpha <- clip(alpha - lr * grad_alpha) with the reference's
    regularized gradient (reference `flow_utils.py:240-250`)."""
    args = client.args
    ga = ops.alpha_grad(client.arena.flat, client.arena_personal.flat,
                        client.arena.grad, client.arena_personal.grad,
                        args.fed_personal_alpha)
    alpha_n = args.fed_personal_alpha - lr * ga
    return float(np.clip(alpha_n, 0.0, 1.0))


def train_and_validate_federated_apfl(client):
    args = client.args
    log('start training and validation with Federated setting.', args.debug)
    if args.evaluate and args.graph.rank == 0:
        do_validate(args, client.model, client.optimizer, client.criterion,
                    client.metrics, client.test_loader, None,
                    data_mode='test')
        return

    tracker = define_local_training_tracker()
    start_global_time = time.time()
    tracker['start_load_time'] = time.time()

    for n_c in range(args.num_comms):
        args.rounds_comm += 1
        args.comm_time.append(0.0)
        log('Starting round {} of training'.format(n_c + 1), args.debug)
        online_clients = client.comm.set_online_clients()
        if n_c == 0 and 0 not in online_clients:
            online_clients = sorted(online_clients + [0])
        online = args.graph.rank in online_clients

        st = time.time()
        distribute_model_server(client.comm, client.model_server)
        client.arena.load_flat(client.model_server)
        args.comm_time[-1] += time.time() - st

        alpha_contrib = (0.0, 0.0)
        gs = getattr(client, 'graph_stepper_apfl', None)
        if gs is None and getattr(args, 'hip_graph', False):
            from fedtorch_amd.trainings.graphstep import GraphStepperAPFL
            gs = client.graph_stepper_apfl = GraphStepperAPFL(client)
        if online:
            is_sync = False
            ep = -1
            while not is_sync:
                ep += 1
                for i, (_input, _target) in enumerate(client.train_loader):
                    client.model.train()
                    client.model_personal.train()
                    logging_load_time(tracker)
                    args.local_index += 1
                    args.local_data_seen += len(_target)
                    get_current_epoch(args)
                    lr = adjust_learning_rate(args, client.optimizer,
                                              client.scheduler)
                    _input, _target = load_data_batch(args, _input, _target,
                                                      tracker)
                    if _input.size(0) == 1:
                        is_sync = is_sync_fed(args)
                        break
                    adaptive_now = (args.fed_adaptive_alpha and i == 0
                                    and ep == 0)
                    stepped = False
                    if gs is not None and gs.ok and not gs.fail:
                        if adaptive_now:
                            # the alpha update reads fresh grads host-
                            # side: take the step on the eager stolen-
                            # gather flow (twin-safe), then update alpha
                            stepped = gs.eager_step(
                                _input, _target, args.fed_personal_alpha)
                            if stepped:
                                args.fed_personal_alpha = \
                                    apfl_alpha_update(client, lr)
                                alpha_contrib = (args.fed_personal_alpha,
                                                 args.graph.n_nodes)
                        else:
                            stepped = gs.maybe_step(
                                _input, _target, lr,
                                args.fed_personal_alpha)
                    if stepped:
                        tracker['start_load_time'] = time.time()
                        is_sync = is_sync_fed(args)
                        if is_sync:
                            break
                        continue
                    # global model step
                    client.optimizer.zero_grad()
                    with amp(args):
                        loss, _ = inference(client.model, client.criterion,
                                            client.metrics, _input, _target)
                    loss.backward()
                    client.optimizer.step(
                        apply_lr=True, apply_in_momentum=args.in_momentum,
                        apply_out_momentum=False)
                    # personal (blended) step
                    client.optimizer.zero_grad()
                    client.optimizer_personal.zero_grad()
                    with amp(args):
                        loss_p, _ = inference_personal(
                            client.model_personal, client.model,
                            args.fed_personal_alpha, client.criterion,
                            client.metrics, _input, _target)
                    loss_p.backward()
                    client.optimizer_personal.step(
                        apply_lr=True, apply_in_momentum=args.in_momentum,
                        apply_out_momentum=False)
                    # adaptive alpha once per round (reference
                    # `apfl.py:119-123`)
                    if args.fed_adaptive_alpha and i == 0 and ep == 0:
                        args.fed_personal_alpha = apfl_alpha_update(client,
                                                                    lr)
                        alpha_contrib = (args.fed_personal_alpha,
                                         args.graph.n_nodes)
                    tracker['start_load_time'] = time.time()
                    is_sync = is_sync_fed(args)
                    if is_sync:
                        break
        else:
            log('Offline in this round. Waiting on others to finish!',
                args.debug)
        if gs is not None:
            gs.flush(tracker)
        if args.fed_adaptive_alpha:
            avg_alpha = global_average(alpha_contrib[0], alpha_contrib[1])
            if online:
                log('New alpha is:{}'.format(float(avg_alpha)), args.debug)

        do_validate(args, client.model, client.optimizer_personal,
                    client.criterion, client.metrics, client.train_loader,
                    None, data_mode='train', personal=True,
                    model_personal=client.model_personal,
                    alpha=args.fed_personal_alpha, skip=not online)
        if args.fed_personal:
            do_validate(args, client.model, client.optimizer_personal,
                        client.criterion, client.metrics, client.val_loader,
                        None, data_mode='validation', personal=True,
                        model_personal=client.model_personal,
                        alpha=args.fed_personal_alpha, skip=not online)

        log('Enter synching', args.debug)
        tracker['start_sync_time'] = time.time()
        args.global_index += 1
        fedavg_aggregation(args, client.comm, client.arena,
                           client.model_server, client.optimizer,
                           online_clients, work=client.work)
        client.comm.flush_comm_time()
        logging_sync_time(tracker)

        do_validate(args, client.model, client.optimizer, client.criterion,
                    client.metrics, client.train_loader, None,
                    data_mode='train', skip=not online)
        if args.fed_personal:
            do_validate(args, client.model, client.optimizer,
                        client.criterion, client.metrics, client.val_loader,
                        None, data_mode='validation', skip=not online)
        logging_globally(tracker, start_global_time)
        start_global_time = time.time()

        if args.fed_personal_test:
            do_validate(args, client.model, client.optimizer_personal,
                        client.criterion, client.metrics, client.test_loader,
                        None, data_mode='test', personal=True,
                        model_personal=client.model_personal,
                        alpha=args.fed_personal_alpha, skip=not online)
        elif args.graph.rank == 0:
            do_validate(args, client.model, client.optimizer,
                        client.criterion, client.metrics, client.test_loader,
                        None, data_mode='test')
        client.comm.barrier()
