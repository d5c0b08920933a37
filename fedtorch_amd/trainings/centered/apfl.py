# -*- coding: utf-8 -*-
"""Centered APFL (parity with reference
`comms/trainings/federated/centered/apfl.py:30-163`)."""
import time

from fedtorch_amd.components.scheduler import adjust_learning_rate
from fedtorch_amd.components.dataset import load_data_batch
from fedtorch_amd.trainings.flow import get_current_epoch, is_sync_fed
from fedtorch_amd.trainings.eval import inference, inference_personal
from fedtorch_amd.trainings.federated import amp
from fedtorch_amd.trainings.apfl import apfl_alpha_update
from fedtorch_amd.trainings.eval_centered import (
    do_validate_centered, log_validation_centered)
from fedtorch_amd.aggregation.centered import (
    fedavg_aggregation_centered, set_online_clients_centered)
from fedtorch_amd.logs.logging import (
    log, logging_sync_time, logging_load_time, logging_globally)
from fedtorch_amd.logs.meter import define_local_training_tracker


def train_and_validate_apfl_centered(Clients, Server, validate=True):
    log('start training and validation of APFL in a centered way.')
    tracker = define_local_training_tracker()
    start_global_time = time.time()
    tracker['start_load_time'] = time.time()
    args = Server.args

    for n_c in range(args.num_comms):
        args.rounds_comm += 1
        args.local_index += 1
        Server.zero_grad()
        Server.reset_tracker(Server.local_val_tracker)
        Server.reset_tracker(Server.global_val_tracker)
        if args.fed_personal:
            Server.reset_tracker(Server.local_personal_val_tracker)
            Server.reset_tracker(Server.global_personal_val_tracker)
        log('Starting round {} of training'.format(n_c + 1))
        online_clients = set_online_clients_centered(args)

        for oc in online_clients:
            C = Clients[oc]
            C.arena.load_flat(Server.arena.flat)
            C.args.rounds_comm = args.rounds_comm
            local_steps = 0
            is_sync = False
            if validate:
                do_validate_centered(C.args, Server.model, Server.criterion,
                                     Server.metrics, Server.optimizer,
                                     C.train_loader, Server.global_val_tracker,
                                     val=False)
            if args.fed_personal:
                if validate:
                    do_validate_centered(C.args, Server.model, Server.criterion,
                                         Server.metrics, Server.optimizer,
                                         C.val_loader,
                                         Server.global_personal_val_tracker,
                                         val=True)
            while not is_sync:
                for _input, _target in C.train_loader:
                    local_steps += 1
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
                    C.optimizer.zero_grad()
                    C.optimizer_personal.zero_grad()
                    with amp(args):
                        loss_p, _ = inference_personal(
                            C.model_personal, C.model,
                            C.args.fed_personal_alpha, C.criterion,
                            C.metrics, _input, _target)
                    loss_p.backward()
                    C.optimizer_personal.step(
                        apply_lr=True, apply_in_momentum=C.args.in_momentum,
                        apply_out_momentum=False)
                    if C.args.fed_adaptive_alpha and local_steps == 1:
                        C.args.fed_personal_alpha = apfl_alpha_update(C, lr)
                        log('New alpha is:{}'.format(
                            C.args.fed_personal_alpha), C.args.debug)
                    tracker['start_load_time'] = time.time()
                    is_sync = is_sync_fed(C.args)
                    if is_sync:
                        break
            if validate:
                do_validate_centered(C.args, C.model, C.criterion, C.metrics,
                                     C.optimizer, C.train_loader,
                                     Server.local_val_tracker, val=False,
                                     personal=True,
                                     model_personal=C.model_personal,
                                     alpha=C.args.fed_personal_alpha)
            if args.fed_personal:
                if validate:
                    do_validate_centered(C.args, C.model, C.criterion, C.metrics,
                                         C.optimizer, C.val_loader,
                                         Server.local_personal_val_tracker,
                                         val=True, personal=True,
                                         model_personal=C.model_personal,
                                         alpha=C.args.fed_personal_alpha)
            tracker['start_sync_time'] = time.time()
            args.global_index += 1
            logging_sync_time(tracker)

        fedavg_aggregation_centered(Clients, Server, online_clients)
        log_validation_centered(args, Server.local_val_tracker, val=False,
                                personal=True)
        if args.fed_personal:
            log_validation_centered(args, Server.local_personal_val_tracker,
                                    val=True, personal=True)
        log_validation_centered(args, Server.global_val_tracker, val=False,
                                local=False)
        if args.fed_personal:
            log_validation_centered(args, Server.global_personal_val_tracker,
                                    val=True, local=False)
        logging_globally(tracker, start_global_time)
        start_global_time = time.time()
