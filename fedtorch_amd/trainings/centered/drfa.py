# -*- coding: utf-8 -*-
"""Centered DRFA (parity with reference
`comms/trainings/federated/centered/drfa.py`): lambda-weighted aggregation
over {fedavg, fedgate, scaffold, qsparse}, kth-model snapshot + average,
dual lambda update on a second sampled set."""
import time

import torch

from fedtorch_amd.components.scheduler import adjust_learning_rate
from fedtorch_amd.components.dataset import load_data_batch
from fedtorch_amd.trainings.flow import get_current_epoch, is_sync_fed
from fedtorch_amd.trainings.eval import inference
from fedtorch_amd.trainings.federated import amp
from fedtorch_amd.trainings.afl import lambda_dual_update
from fedtorch_amd.trainings.eval_centered import (
    do_validate_centered, log_validation_centered, log_test_centered)
from fedtorch_amd.aggregation.centered import (
    fedavg_aggregation_centered, fedgate_aggregation_centered,
    scaffold_aggregation_centered, qsparse_aggregation_centered,
    aggregate_kth_model_centered, set_online_clients_centered)
from fedtorch_amd.logs.logging import (
    log, logging_sync_time, logging_load_time, logging_globally)
from fedtorch_amd.logs.meter import define_local_training_tracker


def train_and_validate_drfa_centered(Clients, Server):
    log('start training and validation of DRFA in a centered way.')
    tracker = define_local_training_tracker()
    start_global_time = time.time()
    tracker['start_load_time'] = time.time()
    args = Server.args

    for oc in range(args.graph.n_nodes):
        Server.lambda_vector[oc] = Clients[oc].args.num_samples_per_epoch
    Server.lambda_vector /= Server.lambda_vector.sum()

    for n_c in range(args.num_comms):
        args.rounds_comm += 1
        args.local_index += 1
        args.drfa_gamma *= 0.9
        Server.zero_grad()
        for tr in (Server.local_val_tracker, Server.global_val_tracker,
                   Server.global_test_tracker):
            Server.reset_tracker(tr)
        if args.fed_personal:
            Server.reset_tracker(Server.local_personal_val_tracker)
            Server.reset_tracker(Server.global_personal_val_tracker)
        log('Starting round {} of training'.format(n_c + 1))
        online_clients = set_online_clients_centered(args)
        k = int(torch.randint(low=1, high=max(args.local_step, 2),
                              size=(1,)))
        local_steps, lr = 0, args.old_learning_rate

        for oc in online_clients:
            C = Clients[oc]
            C.arena.load_flat(Server.arena.flat)
            C.args.rounds_comm = args.rounds_comm
            local_steps = 0
            is_sync = False
            t = args.federated_type
            if t == 'fedgate':
                C.optimizer.set_correction(delta=C.model_delta)
            elif t == 'scaffold':
                C.optimizer.set_correction(
                    ctrl_server=Server.model_server_control,
                    ctrl_client=C.model_client_control)
            else:
                C.optimizer.clear_correction()
            while not is_sync:
                for _input, _target in C.train_loader:
                    local_steps += 1
                    if k == local_steps:
                        C.kth_model.copy_(C.arena.flat)
                    C.model.train()
                    logging_load_time(tracker)
                    C.args.local_index += 1
                    C.args.local_data_seen += len(_target)
                    get_current_epoch(C.args)
                    lr = adjust_learning_rate(C.args, C.optimizer,
                                              C.scheduler)
                    _input, _target = load_data_batch(C.args, _input,
                                                      _target, tracker)
                    if _input.size(0) == 1:
                        is_sync = is_sync_fed(C.args)
                        break
                    C.optimizer.zero_grad()
                    with amp(args):
                        loss, _ = inference(C.model, C.criterion, C.metrics,
                                            _input, _target)
                    loss.backward()
                    C.optimizer.step(apply_lr=True,
                                     apply_in_momentum=C.args.in_momentum,
                                     apply_out_momentum=False)
                    tracker['start_load_time'] = time.time()
                    is_sync = is_sync_fed(C.args)
                    if is_sync:
                        break
            do_validate_centered(C.args, C.model, C.criterion, C.metrics,
                                 C.optimizer, C.train_loader,
                                 Server.local_val_tracker, val=False,
                                 local=True)
            tracker['start_sync_time'] = time.time()
            args.global_index += 1
            logging_sync_time(tracker)

        t = args.federated_type
        lam = Server.lambda_vector.numpy()
        if t == 'scaffold':
            scaffold_aggregation_centered(Clients, Server, online_clients,
                                          local_steps, lr, lambda_weight=lam)
        elif t == 'fedgate':
            fedgate_aggregation_centered(Clients, Server, online_clients,
                                         local_steps, lr, lambda_weight=lam)
        elif t == 'qsparse':
            qsparse_aggregation_centered(Clients, Server, online_clients,
                                         local_steps, lr, lambda_weight=lam)
        else:
            fedavg_aggregation_centered(Clients, Server, online_clients,
                                        lambda_weight=lam)
        aggregate_kth_model_centered(Clients, Server, online_clients)

        # dual update on a second sampled set (reference `drfa.py:235-253`)
        online_clients_lambda = set_online_clients_centered(args)
        loss_tensor = torch.zeros(args.graph.n_nodes)
        for ocl in online_clients_lambda:
            C = Clients[ocl]
            saved = C.arena.clone_flat()
            C.arena.load_flat(Server.kth_model)
            for _input, _target in C.train_loader:
                _input, _target = load_data_batch(C.args, _input, _target,
                                                  tracker)
                if _input.size(0) == 1:
                    break
                C.model.eval()
                with torch.no_grad(), amp(args):
                    loss, _ = inference(C.model, C.criterion, C.metrics,
                                        _input, _target)
                C.model.train()
                loss_tensor[ocl] = loss.item() * (
                    args.graph.n_nodes / len(online_clients_lambda))
                break
            C.arena.load_flat(saved)
        Server.lambda_vector = lambda_dual_update(
            args, Server.lambda_vector, loss_tensor,
            step_scale=args.local_step)

        log_validation_centered(args, Server.local_val_tracker, val=False,
                                local=True)
        do_validate_centered(args, Server.model, Server.criterion,
                             Server.metrics, Server.optimizer,
                             Server.test_loader, Server.global_test_tracker,
                             val=False, local=False)
        log_test_centered(args, Server.global_test_tracker)
        logging_globally(tracker, start_global_time)
        start_global_time = time.time()
