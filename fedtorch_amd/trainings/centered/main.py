# -*- coding: utf-8 -*-
"""Centered federated round loop (parity with reference
`comms/trainings/federated/centered/main.py:29-244`): sequential loop over
online virtual clients, all resident on one GPU.

Supports fedavg / fedprox / scaffold / fedgate / qsparse / fedadam / qffl /
perfedavg.
"""
import time

import torch

from fedtorch_amd.components.scheduler import adjust_learning_rate
from fedtorch_amd.components.dataset import load_data_batch
from fedtorch_amd.trainings.flow import get_current_epoch, is_sync_fed
from fedtorch_amd.trainings.eval import inference
from fedtorch_amd.trainings.federated import amp
from fedtorch_amd.trainings.eval_centered import (
    do_validate_centered, log_validation_centered,
    log_validation_per_client_centered, log_test_centered)
from fedtorch_amd.aggregation.centered import (
    fedavg_aggregation_centered, fedgate_aggregation_centered,
    scaffold_aggregation_centered, qsparse_aggregation_centered,
    qffl_aggregation_centered, set_online_clients_centered)
from fedtorch_amd.logs.logging import (
    log, logging_sync_time, logging_load_time, logging_globally)
from fedtorch_amd.logs.meter import define_local_training_tracker


def run_client_local_steps(client, Server, tracker, val_batch=None):
    """tau local steps of one centered client (reference
    `centered/main.py:95-178`)."""
    args = client.args
    local_steps = 0
    lr = args.old_learning_rate
    is_sync = False
    t = args.federated_type
    if t == 'fedgate':
        client.optimizer.set_correction(delta=client.model_delta)
    elif t == 'scaffold':
        client.optimizer.set_correction(
            ctrl_server=Server.model_server_control,
            ctrl_client=client.model_client_control)
    elif t == 'fedprox':
        client.optimizer.set_correction(prox_mu=args.fedprox_mu,
                                        server=Server.arena.flat)
    else:
        client.optimizer.clear_correction()
    while not is_sync:
        if args.arch == 'rnn':
            client.model.init_hidden(args.batch_size)
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
                is_sync = is_sync_fed(args)
                break
            client.optimizer.zero_grad()
            with amp(args):
                loss, _ = inference(client.model, client.criterion,
                                    client.metrics, _input, _target,
                                    rnn=args.arch == 'rnn')
            loss.backward()
            if 'robust' in args.arch:
                client.model.noise.grad.data *= -1
            client.optimizer.step(apply_lr=True,
                                  apply_in_momentum=args.in_momentum,
                                  apply_out_momentum=False)
            if 'robust' in args.arch:
                nrm = torch.norm(client.model.noise.data)
                if nrm > 1:
                    client.model.noise.data.div_(nrm)
            if t == 'perfedavg' and val_batch is not None:
                # lookahead second step at beta (reference
                # `centered/main.py:156-170`)
                _iv, _tv = val_batch
                lr = adjust_learning_rate(args, client.optimizer,
                                          client.scheduler,
                                          lr_external=args.perfedavg_beta)
                if _iv.size(0) == 1:
                    is_sync = is_sync_fed(args)
                    break
                client.optimizer.zero_grad()
                with amp(args):
                    loss, _ = inference(client.model, client.criterion,
                                        client.metrics, _iv, _tv)
                loss.backward()
                client.optimizer.step(apply_lr=True,
                                      apply_in_momentum=args.in_momentum,
                                      apply_out_momentum=False)
            tracker['start_load_time'] = time.time()
            is_sync = is_sync_fed(args)
            if is_sync:
                break
    return local_steps, lr


def train_and_validate_federated_centered(Clients, Server, validate=True):
    log('start training and validation with Federated setting in a '
        'centered way.')
    tracker = define_local_training_tracker()
    start_global_time = time.time()
    tracker['start_load_time'] = time.time()
    args = Server.args

    for n_c in range(args.num_comms):
        args.rounds_comm += 1
        args.local_index += 1
        Server.zero_grad()
        for tr in (Server.local_val_tracker, Server.global_val_tracker,
                   Server.global_test_tracker):
            Server.reset_tracker(tr)
        if args.fed_personal:
            Server.reset_tracker(Server.local_personal_val_tracker)
            Server.reset_tracker(Server.global_personal_val_tracker)

        log('Starting round {} of training'.format(n_c + 1))
        online_clients = set_online_clients_centered(args)
        local_steps, lr = 0, args.old_learning_rate
        for oc in online_clients:
            Clients[oc].arena.load_flat(Server.arena.flat)
            if Server.arena.buf_flat is not None:
                Clients[oc].arena.buf_flat.copy_(Server.arena.buf_flat)
            Clients[oc].args.rounds_comm = args.rounds_comm

            if args.federated_type == 'qffl':
                Clients[oc].full_loss = 0.0
                for _input, _target in Clients[oc].train_loader:
                    _input, _target = load_data_batch(
                        Clients[oc].args, _input, _target, tracker)
                    if _input.size(0) == 1:
                        break
                    with torch.no_grad(), amp(args):
                        loss, _ = inference(
                            Clients[oc].model, Clients[oc].criterion,
                            Clients[oc].metrics, _input, _target,
                            rnn=args.arch == 'rnn')
                    Clients[oc].full_loss += loss.item()

            # pre-training validation of the server model on client data
            if validate:
                do_validate_centered(
                    Clients[oc].args, Server.model, Server.criterion,
                    Server.metrics, Server.optimizer,
                    Clients[oc].train_loader,
                    Server.global_val_tracker, val=False, local=False)
            if validate and args.per_class_acc:
                # per-client trackers feed the worst/best/var lines
                # (reference `centered/main.py:77-88`)
                Clients[oc].reset_tracker(Clients[oc].local_val_tracker)
                Clients[oc].reset_tracker(Clients[oc].global_val_tracker)
                do_validate_centered(
                    Clients[oc].args, Server.model, Server.criterion,
                    Server.metrics, Server.optimizer,
                    Clients[oc].train_loader,
                    Clients[oc].global_val_tracker, val=False, local=False)
            if validate and args.fed_personal:
                do_validate_centered(
                    Clients[oc].args, Server.model, Server.criterion,
                    Server.metrics, Server.optimizer,
                    Clients[oc].val_loader,
                    Server.global_personal_val_tracker, val=True,
                    local=False)

            val_batch = None
            if args.federated_type == 'perfedavg':
                for _iv, _tv in Clients[oc].val_loader1:
                    _iv, _tv = load_data_batch(Clients[oc].args, _iv, _tv,
                                               tracker)
                    val_batch = (_iv, _tv)
                    break

            local_steps, lr = run_client_local_steps(
                Clients[oc], Server, tracker, val_batch=val_batch)

            # post-training validation of the client model
            if validate:
                do_validate_centered(
                    Clients[oc].args, Clients[oc].model,
                    Clients[oc].criterion, Clients[oc].metrics,
                    Clients[oc].optimizer,
                    Clients[oc].train_loader, Server.local_val_tracker,
                    val=False, local=True)
            if validate and args.per_class_acc:
                do_validate_centered(
                    Clients[oc].args, Clients[oc].model,
                    Clients[oc].criterion, Clients[oc].metrics,
                    Clients[oc].optimizer, Clients[oc].train_loader,
                    Clients[oc].local_val_tracker, val=False, local=True)
            if validate and args.fed_personal:
                do_validate_centered(
                    Clients[oc].args, Clients[oc].model,
                    Clients[oc].criterion, Clients[oc].metrics,
                    Clients[oc].optimizer, Clients[oc].val_loader,
                    Server.local_personal_val_tracker, val=True, local=True)
            tracker['start_sync_time'] = time.time()
            args.global_index += 1
            logging_sync_time(tracker)

        t = args.federated_type
        if t == 'scaffold':
            scaffold_aggregation_centered(Clients, Server, online_clients,
                                          local_steps, lr)
        elif t == 'fedgate':
            fedgate_aggregation_centered(Clients, Server, online_clients,
                                         local_steps, lr)
        elif t == 'qsparse':
            qsparse_aggregation_centered(Clients, Server, online_clients,
                                         local_steps, lr)
        elif t == 'qffl':
            qffl_aggregation_centered(Clients, Server, online_clients, lr)
        else:
            fedavg_aggregation_centered(Clients, Server, online_clients)
        from fedtorch_amd.aggregation.federated import \
            aggregate_bn_buffers_centered
        aggregate_bn_buffers_centered(Clients, Server, online_clients)

        log_validation_centered(args, Server.local_val_tracker, val=False,
                                local=True)
        log_validation_centered(args, Server.global_val_tracker, val=False,
                                local=False)
        if args.fed_personal:
            log_validation_centered(args, Server.local_personal_val_tracker,
                                    val=True, local=True)
            log_validation_centered(args, Server.global_personal_val_tracker,
                                    val=True, local=False)
        if args.per_class_acc:
            log_validation_per_client_centered(args, Clients, online_clients,
                                               val=False, local=False)
            log_validation_per_client_centered(args, Clients, online_clients,
                                               val=False, local=True)

        if validate:
            do_validate_centered(args, Server.model, Server.criterion,
                                 Server.metrics, Server.optimizer,
                                 Server.test_loader,
                                 Server.global_test_tracker,
                                 val=False, local=False)
            log_test_centered(args, Server.global_test_tracker)
        logging_globally(tracker, start_global_time)
        start_global_time = time.time()
