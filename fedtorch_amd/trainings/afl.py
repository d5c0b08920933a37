# -*- coding: utf-8 -*-
"""AFL — Agnostic Federated Learning (arXiv:1902.00146); parity with
reference `comms/trainings/federated/afl.py:32-186` and
`comms/algorithms/federated/afl.py:9-60`.

Aggregation weight is lambda_i directly (not 1/K); the dual lambda update
(+ simplex projection + 1e-3 floor) runs on rank 0 from an all-gathered
per-client loss vector.
"""
import time

import torch

from fedtorch_amd import ops
from fedtorch_amd.components.scheduler import adjust_learning_rate
from fedtorch_amd.components.dataset import load_data_batch
from fedtorch_amd.trainings.flow import get_current_epoch, is_sync_fed
from fedtorch_amd.trainings.eval import inference, do_validate
from fedtorch_amd.trainings.federated import amp
from fedtorch_amd.aggregation.federated import (distribute_model_server,
                                                _buf)
from fedtorch_amd.logs.logging import (
    log, logging_sync_time, logging_load_time, logging_globally)
from fedtorch_amd.logs.meter import define_local_training_tracker


def afl_aggregation(args, comm, arena, server_flat, optimizer,
                    online_clients, lambda_weight, loss, work):
    """lambda-weighted diff all-reduce + loss all-gather (reference
    `algorithms/federated/afl.py:9-60`)."""
    rank = args.graph.rank
    if rank in online_clients:
        w = float(lambda_weight)
    else:
        w = 0.0
    agg = _buf(work, 'agg', arena.flat)
    ops.weighted_diff_restore(server_flat, arena.flat, agg, w)
    comm.all_reduce(agg)
    loss_tensor = comm.gather_scalar(loss)
    optimizer.step(apply_lr=False, scale=args.lr_scale_at_sync,
                   apply_in_momentum=False,
                   apply_out_momentum=args.out_momentum, grad=agg)
    server_flat.copy_(arena.flat)
    return loss_tensor


def lambda_dual_update(args, lambda_vector, loss_tensor, step_scale=1.0):
    """lambda += gamma*scale*loss; simplex projection; 1e-3 floor
    (reference `afl.py:158-170`, `drfa.py:242-249`)."""
    lambda_vector += args.drfa_gamma * step_scale * loss_tensor
    lambda_vector = ops.euclidean_proj_simplex(lambda_vector)
    zeros = lambda_vector <= 1e-3
    if zeros.sum() > 0:
        lambda_vector[zeros] = 1e-3
        lambda_vector /= lambda_vector.sum()
    return lambda_vector


def train_and_validate_federated_afl(client):
    args = client.args
    log('start training and validation with Federated setting.', args.debug)
    if args.evaluate and args.graph.rank == 0:
        do_validate(args, client.model, client.optimizer, client.criterion,
                    client.metrics, client.test_loader, None,
                    data_mode='test')
        return

    # lambda init proportional to sample sizes (reference `afl.py:46-51`);
    # all-gather lets every rank hold the true vector.
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

        st = time.time()
        distribute_model_server(client.comm, client.model_server)
        client.comm.broadcast(client.lambda_vector, src=0)
        client.arena.load_flat(client.model_server)
        args.comm_time[-1] += time.time() - st

        loss = torch.tensor(0.0)
        gs = getattr(client, 'graph_stepper', None)
        if gs is None and getattr(args, 'hip_graph', False):
            from fedtorch_amd.trainings.graphstep import GraphStepper
            gs = client.graph_stepper = GraphStepper(client)
        if online:
            is_sync = False
            while not is_sync:
                for _input, _target in client.train_loader:
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
                        if gs.last_loss() is not None:
                            loss = gs.last_loss()
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
        loss_tensor = afl_aggregation(
            args, client.comm, client.arena, client.model_server,
            client.optimizer, online_clients,
            client.lambda_vector[args.graph.rank].item(),
            float(loss.item()) if online else 0.0, client.work)
        client.comm.flush_comm_time()
        logging_sync_time(tracker)

        do_validate(args, client.model, client.optimizer, client.criterion,
                    client.metrics, client.train_loader, None,
                    data_mode='train', skip=not online)
        if args.fed_personal:
            do_validate(args, client.model, client.optimizer,
                        client.criterion, client.metrics, client.val_loader,
                        None, data_mode='validation', skip=not online)

        # dual update on rank 0 (reference `afl.py:158-170`)
        if args.graph.rank == 0:
            client.lambda_vector = lambda_dual_update(
                args, client.lambda_vector, loss_tensor)

        logging_globally(tracker, start_global_time)
        start_global_time = time.time()
        if args.graph.rank == 0:
            do_validate(args, client.model, client.optimizer,
                        client.criterion, client.metrics, client.test_loader,
                        None, data_mode='test')
        log('This round communication time is: {}'.format(
            args.comm_time[-1]), args.debug)
        client.comm.barrier()
