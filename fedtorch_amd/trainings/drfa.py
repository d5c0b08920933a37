# -*- coding: utf-8 -*-
"""DRFA — Distributionally Robust Federated Averaging (NeurIPS 2020);
parity with reference `comms/trainings/federated/drfa.py:38-259`.

Meta-algorithm over {fedavg, fedgate, scaffold}: lambda-weighted
aggregation, a kth-model snapshot averaged across clients, and a dual
lambda update on a second sampled client set.
"""
import time

import torch

from fedtorch_amd.components.scheduler import adjust_learning_rate
from fedtorch_amd.components.dataset import load_data_batch
from fedtorch_amd.trainings.flow import get_current_epoch, is_sync_fed
from fedtorch_amd.trainings.eval import inference, do_validate
from fedtorch_amd.trainings.federated import (amp, set_round_correction,
                                              aggregate_round)
from fedtorch_amd.trainings.afl import lambda_dual_update
from fedtorch_amd.aggregation.federated import (
    distribute_model_server, distribute_model_server_control,
    aggregate_models_virtual)
from fedtorch_amd.logs.logging import (
    log, logging_sync_time, logging_load_time, logging_globally)
from fedtorch_amd.logs.meter import define_local_training_tracker


def train_and_validate_federated_drfa(client):
    args = client.args
    log('start DRFA training (aggregator: {}).'.format(args.federated_type),
        args.debug)
    if args.evaluate and args.graph.rank == 0:
        do_validate(args, client.model, client.optimizer, client.criterion,
                    client.metrics, client.test_loader, None,
                    data_mode='test')
        return

    # lambda init proportional to sample sizes (reference `drfa.py:51-57`)
    sizes = client.comm.gather_scalar(args.num_samples_per_epoch)
    client.lambda_vector = sizes / float(args.train_dataset_size)

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
        args.drfa_gamma *= 0.9

        st = time.time()
        if args.federated_type == 'scaffold':
            distribute_model_server_control(
                client.comm, client.model_server,
                client.model_server_control, client.work)
        else:
            distribute_model_server(client.comm, client.model_server)
        client.arena.load_flat(client.model_server)
        # lambda + random cut index k (reference `drfa.py:94-98`)
        client.comm.broadcast(client.lambda_vector, src=0)
        k = torch.randint(low=1, high=max(args.local_step, 2), size=(1,))
        client.comm.broadcast(k, src=0)
        args.comm_time[-1] += time.time() - st
        k = int(k[0])

        set_round_correction(client)
        gs = getattr(client, 'graph_stepper', None)
        if gs is None and getattr(args, 'hip_graph', False):
            from fedtorch_amd.trainings.graphstep import GraphStepper
            gs = client.graph_stepper = GraphStepper(client)
        local_steps = 0
        lr = args.old_learning_rate
        if online:
            is_sync = False
            while not is_sync:
                for _input, _target in client.train_loader:
                    local_steps += 1
                    if k == local_steps:
                        client.kth_model.copy_(client.arena.flat)
                    client.model.train()
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
                    if gs is not None and gs.maybe_step(_input, _target,
                                                        lr):
                        # the kth snapshot happens BETWEEN replays (above)
                        # so the graph path preserves DRFA semantics
                        tracker['start_load_time'] = time.time()
                        is_sync = is_sync_fed(args)
                        if is_sync:
                            break
                        continue
                    client.optimizer.zero_grad()
                    with amp(args):
                        loss, _ = inference(client.model, client.criterion,
                                            client.metrics, _input, _target)
                    loss.backward()
                    client.optimizer.step(
                        apply_lr=True, apply_in_momentum=args.in_momentum,
                        apply_out_momentum=False)
                    tracker['start_load_time'] = time.time()
                    is_sync = is_sync_fed(args)
                    if is_sync:
                        break
        else:
            log('Offline in this round. Waiting on others to finish!',
                args.debug)
        if gs is not None:
            gs.flush(tracker)

        do_validate(args, client.model, client.optimizer, client.criterion,
                    client.metrics, client.train_loader, None,
                    data_mode='train', local=True, skip=not online)
        if args.fed_personal:
            do_validate(args, client.model, client.optimizer,
                        client.criterion, client.metrics, client.val_loader,
                        None, data_mode='validation', local=True,
                        skip=not online)

        log('Enter synching', args.debug)
        tracker['start_sync_time'] = time.time()
        args.global_index += 1
        aggregate_round(client, online_clients, lr, local_steps,
                        lambda_weight=client.lambda_vector[
                            args.graph.rank].item())
        # average the kth model (reference `drfa.py:195`)
        client.kth_model.copy_(aggregate_models_virtual(
            args, client.comm, client.kth_model, online_clients,
            work=client.work))
        client.comm.flush_comm_time()
        logging_sync_time(tracker)

        do_validate(args, client.model, client.optimizer, client.criterion,
                    client.metrics, client.train_loader, None,
                    data_mode='train', skip=not online)
        if args.fed_personal:
            do_validate(args, client.model, client.optimizer,
                        client.criterion, client.metrics, client.val_loader,
                        None, data_mode='validation', skip=not online)
        if args.graph.rank == 0:
            do_validate(args, client.model, client.optimizer,
                        client.criterion, client.metrics, client.test_loader,
                        None, data_mode='test')

        # ---- lambda update on a second sampled set (reference
        # `drfa.py:216-248`) ----
        online_clients_lambda = client.comm.set_online_clients()
        online_lambda = args.graph.rank in online_clients_lambda
        st = time.time()
        distribute_model_server(client.comm, client.kth_model)
        args.comm_time[-1] += time.time() - st
        loss = torch.tensor(0.0)
        if online_lambda:
            # one-batch loss of the kth model: swap arenas, eval, restore
            saved = client.arena.clone_flat()
            client.arena.load_flat(client.kth_model)
            for _input, _target in client.train_loader:
                _input, _target = load_data_batch(args, _input, _target,
                                                  tracker)
                if _input.size(0) == 1:
                    break
                client.model.eval()
                with torch.no_grad(), amp(args):
                    loss, _ = inference(client.model, client.criterion,
                                        client.metrics, _input, _target)
                client.model.train()
                break
            client.arena.load_flat(saved)
        loss_tensor = client.comm.gather_scalar(
            float(loss.item()) if online_lambda else 0.0)
        if args.graph.rank == 0:
            os_lambda = sorted(set(online_clients_lambda) | {0})
            num_online = len(os_lambda)
            # scale the sampled losses by n/K (reference `drfa.py:238`)
            loss_tensor = loss_tensor * (args.graph.n_nodes / num_online)
            client.lambda_vector = lambda_dual_update(
                args, client.lambda_vector, loss_tensor,
                step_scale=args.local_step)

        logging_globally(tracker, start_global_time)
        start_global_time = time.time()
        log('This round communication time is: {}'.format(
            args.comm_time[-1]), args.debug)
        client.comm.barrier()
