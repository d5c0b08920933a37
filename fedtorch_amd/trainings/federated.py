# -*- coding: utf-8 -*-
"""Federated round loop (parity with reference
`comms/trainings/federated/main.py:34-212`): FedAvg / FedProx / SCAFFOLD /
FedGATE / Qsparse / FedAdam.

MI355X-native structure: the server model and all aux state are flat arena
buffers; model distribution is ONE broadcast, aggregation ONE weighted
collective (`fedtorch_amd/aggregation/federated.py`); per-algorithm gradient
corrections are fused into the local-SGD step kernel
(`optimizer.set_correction`) instead of per-parameter Python loops
(reference `main.py:116-129`).
"""
import time
from contextlib import nullcontext

import torch

from fedtorch_amd.components.scheduler import adjust_learning_rate
from fedtorch_amd.components.dataset import load_data_batch
from fedtorch_amd.trainings.flow import get_current_epoch, is_sync_fed
from fedtorch_amd.trainings.eval import inference, do_validate
from fedtorch_amd.aggregation.federated import (
    fedavg_aggregation, fedgate_aggregation, scaffold_aggregation,
    qsparse_aggregation, distribute_model_server,
    distribute_model_server_control)
from fedtorch_amd.logs.logging import (
    log, logging_sync_time, logging_load_time, logging_globally)
from fedtorch_amd.logs.meter import define_local_training_tracker


def amp(args):
    """bf16 autocast context (fp32 master weights stay in the arena)."""
    if getattr(args, 'bf16', False) and args.graph.on_cuda:
        return torch.autocast('cuda', dtype=torch.bfloat16)
    return nullcontext()


def set_round_correction(client):
    """Arm the fused per-step gradient correction for this round
    (reference applies these per-parameter per-step, `main.py:116-129`)."""
    t = client.args.federated_type
    if t == 'fedgate':
        client.optimizer.set_correction(delta=client.model_delta)
    elif t == 'scaffold':
        client.optimizer.set_correction(
            ctrl_server=client.model_server_control,
            ctrl_client=client.model_client_control)
    elif t == 'fedprox':
        client.optimizer.set_correction(
            prox_mu=client.args.fedprox_mu, server=client.model_server)
    else:
        client.optimizer.clear_correction()


def run_local_steps(client, tracker, online=True, lambda_weight=None):
    """tau local steps until the sync predicate fires (reference
    `main.py:83-158`). Returns (local_steps, lr)."""
    args = client.args
    local_steps = 0
    lr = args.old_learning_rate
    if not online:
        return local_steps, lr
    gs = getattr(client, 'graph_stepper', None)
    if gs is None and getattr(args, 'hip_graph', False):
        from fedtorch_amd.trainings.graphstep import GraphStepper
        gs = client.graph_stepper = GraphStepper(client)
    is_sync = False
    while not is_sync:
        for _input, _target in client.train_loader:
            local_steps += 1
            client.model.train()
            logging_load_time(tracker)
            args.local_index += 1
            args.local_data_seen += len(_target)
            get_current_epoch(args)
            lr = adjust_learning_rate(args, client.optimizer,
                                      client.scheduler)
            _input, _target = load_data_batch(args, _input, _target, tracker)
            if _input.size(0) == 1:
                # BatchNorm can't take size-1 batches (reference
                # `main.py:104-106`)
                is_sync = is_sync_fed(args)
                break
            if gs is not None and gs.maybe_step(_input, _target, lr):
                # hipGraph replay did fwd/bwd/fused step; metrics land in
                # the tracker at the sync flush
                is_sync = is_sync_fed(args)
                if is_sync:
                    break
                continue
            client.optimizer.zero_grad()
            with amp(args):
                loss, performance = inference(
                    client.model, client.criterion, client.metrics,
                    _input, _target, rnn=args.arch == 'rnn')
            loss.backward()
            if 'robust' in args.arch:
                client.model.noise.grad.data *= -1  # ascent on the noise
            client.optimizer.step(
                apply_lr=True,
                apply_in_momentum=args.in_momentum,
                apply_out_momentum=False)
            if 'robust' in args.arch:
                nrm = torch.norm(client.model.noise.data)
                if nrm > 1:
                    client.model.noise.data.div_(nrm)
            is_sync = is_sync_fed(args)
            if is_sync:
                break
    if gs is not None:
        gs.flush(tracker)
    return local_steps, lr


def aggregate_round(client, online_clients, lr, local_steps,
                    lambda_weight=None):
    """Dispatch the per-type aggregation (reference `main.py:171-187`)."""
    args = client.args
    t = args.federated_type
    if t == 'fedgate':
        fedgate_aggregation(
            args, client.comm, client.arena, client.model_server,
            client.model_delta, client.model_memory, client.optimizer,
            online_clients, lr, local_steps, lambda_weight=lambda_weight,
            work=client.work)
    elif t == 'scaffold':
        scaffold_aggregation(
            args, client.comm, client.arena, client.model_server,
            client.model_server_control, client.model_client_control,
            client.optimizer, online_clients, lr, local_steps,
            lambda_weight=lambda_weight, work=client.work)
    elif t == 'qsparse':
        qsparse_aggregation(
            args, client.comm, client.arena, client.model_server,
            client.model_memory, client.optimizer, online_clients,
            lambda_weight=lambda_weight, work=client.work)
    else:
        fedavg_aggregation(
            args, client.comm, client.arena, client.model_server,
            client.optimizer, online_clients, lambda_weight=lambda_weight,
            work=client.work)
    from fedtorch_amd.aggregation.federated import aggregate_bn_buffers
    aggregate_bn_buffers(args, client.comm, client.arena, online_clients,
                         work=client.work)


def train_and_validate_federated(client, validate=True):
    """The main federated round loop (reference `main.py:34-212`)."""
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
    log('enter the training.', args.debug)

    for n_c in range(args.num_comms):
        args.rounds_comm += 1
        args.comm_time.append(0.0)
        log('Starting round {} of training'.format(n_c + 1), args.debug)
        online_clients = client.comm.set_online_clients()
        if n_c == 0 and 0 not in online_clients:
            # first round forces the server online (reference `main.py:62-63`)
            online_clients = sorted(online_clients + [0])
        online = args.graph.rank in online_clients

        # model distribution: ONE arena broadcast (scaffold: 2N).
        st = time.time()
        if args.federated_type == 'scaffold':
            distribute_model_server_control(
                client.comm, client.model_server,
                client.model_server_control, client.work)
        else:
            distribute_model_server(client.comm, client.model_server)
        client.arena.load_flat(client.model_server)
        args.comm_time[-1] += time.time() - st

        set_round_correction(client)
        local_steps, lr = run_local_steps(client, tracker, online=online)
        if not online:
            log('Offline in this round. Waiting on others to finish!',
                args.debug)

        # pre-sync validation on local models (reference `main.py:161-165`)
        if validate:
            do_validate(args, client.model, client.optimizer,
                        client.criterion, client.metrics,
                        client.train_loader, None, data_mode='train',
                        local=True, skip=not online)
            if args.fed_personal:
                do_validate(args, client.model, client.optimizer,
                            client.criterion, client.metrics,
                            client.val_loader, None, data_mode='validation',
                            local=True, skip=not online)

        log('Enter synching', args.debug)
        tracker['start_sync_time'] = time.time()
        args.global_index += 1
        if args.check_model_at_sync:
            from fedtorch_amd.logs.check_training import check_model_at_sync
            check_model_at_sync(args, client.arena, tag='pre-sync')
        if args.track_model_aggregation:
            local_flat_before = client.arena.clone_flat()
            server_before = client.model_server.clone()
        aggregate_round(client, online_clients, lr, local_steps)
        if args.track_model_aggregation:
            from fedtorch_amd.logs.check_training import \
                track_model_aggregation
            track_model_aggregation(
                args,
                local_diff_flat=local_flat_before - server_before,
                agg_diff_flat=client.arena.flat - server_before,
                init_flat=client.work.setdefault(
                    'init_flat', server_before.clone()),
                current_flat=client.arena.flat)
        client.comm.flush_comm_time()
        logging_sync_time(tracker)

        # post-sync validation on the server model (reference `main.py:192-196`)
        if validate:
            do_validate(args, client.model, client.optimizer,
                        client.criterion, client.metrics,
                        client.train_loader, None, data_mode='train',
                        skip=not online)
            if args.fed_personal:
                do_validate(args, client.model, client.optimizer,
                            client.criterion, client.metrics,
                            client.val_loader, None, data_mode='validation',
                            skip=not online)

        logging_globally(tracker, start_global_time)
        start_global_time = time.time()

        if validate and args.graph.rank == 0:
            do_validate(args, client.model, client.optimizer,
                        client.criterion, client.metrics,
                        client.test_loader, None, data_mode='test')
        log('This round communication time is: {}'.format(
            args.comm_time[-1]), args.debug)
        client.comm.barrier()
