# -*- coding: utf-8 -*-
"""Distributed local SGD (parity with reference
`comms/trainings/distributed.py:23-134`): tau local steps between model-diff
all-reduces, arena-fused."""
import gc
import time

from fedtorch_amd.components.scheduler import adjust_learning_rate
from fedtorch_amd.components.dataset import load_data_batch
from fedtorch_amd.trainings.flow import (
    get_current_epoch, get_current_local_step, is_stop)
from fedtorch_amd.trainings.eval import inference, do_validate
from fedtorch_amd.trainings.federated import amp
from fedtorch_amd.aggregation.distributed import aggregate_gradients
from fedtorch_amd.aggregation.federated import _buf
from fedtorch_amd.logs.logging import (
    log, logging_computing, logging_sync_time, logging_display_training,
    logging_load_time, logging_globally)
from fedtorch_amd.logs.meter import define_local_training_tracker


def train_and_validate(client):
    args = client.args
    log('start training and validation.', args.debug)

    if args.evaluate and args.graph.rank == 0:
        do_validate(args, client.model, client.optimizer, client.criterion,
                    client.metrics, client.test_loader, None,
                    data_mode='test')
        return

    tracker = define_local_training_tracker()
    start_global_time = time.time()
    tracker['start_load_time'] = time.time()
    log('enter the training.', args.debug)
    args.comm_time.append(0.0)
    gs = getattr(client, 'graph_stepper', None)
    if gs is None and getattr(args, 'hip_graph', False):
        from fedtorch_amd.trainings.graphstep import GraphStepper
        gs = client.graph_stepper = GraphStepper(client)

    while True:
        for _input, _target in client.train_loader:
            client.model.train()
            logging_load_time(tracker)
            args.local_index += 1
            args.local_data_seen += len(_target)
            get_current_epoch(args)
            local_step = get_current_local_step(args)
            lr = adjust_learning_rate(args, client.optimizer,
                                      client.scheduler)
            _input, _target = load_data_batch(args, _input, _target, tracker)
            if gs is not None and _input.size(0) > 1 and \
                    gs.maybe_step(_input, _target, lr):
                # hipGraph replay; metrics land at the sync flush — keep
                # the timing bookkeeping logging_computing would have set
                tracker['start_sync_time'] = time.time()
                tracker['start_load_time'] = time.time()
            else:
                client.optimizer.zero_grad()
                with amp(args):
                    loss, performance = inference(
                        client.model, client.criterion, client.metrics,
                        _input, _target, rnn=args.arch == 'rnn')
                loss.backward()
                client.optimizer.step(
                    apply_lr=True,
                    apply_in_momentum=args.in_momentum,
                    apply_out_momentum=False)
                logging_computing(tracker, loss, performance, _input, lr)

            is_sync = args.local_index % local_step == 0
            if args.epoch_ % 1 == 0:
                args.finish_one_epoch = True

            if is_sync:
                log('Enter synching', args.debug)
                if gs is not None:
                    gs.flush(tracker, lr=lr)
                args.global_index += 1
                aggregate_gradients(args, client.comm, client.arena,
                                    client.model_server, client.optimizer,
                                    _buf(client.work, 'agg',
                                         client.arena.flat))
                client.comm.flush_comm_time()
                logging_sync_time(tracker)
                logging_globally(tracker, start_global_time)
                start_global_time = time.time()

            if args.finish_one_epoch:
                if gs is not None:
                    gs.flush(tracker, lr=lr)
                if args.epoch % args.eval_freq == 0 and \
                        args.graph.rank == 0:
                    do_validate(args, client.model, client.optimizer,
                                client.criterion, client.metrics,
                                client.test_loader, None, data_mode='test')
                client.comm.barrier()
                args.finish_one_epoch = False
                tracker = define_local_training_tracker()

            if is_stop(args):
                log('Enter final synching', args.debug)
                args.global_index += 1
                aggregate_gradients(args, client.comm, client.arena,
                                    client.model_server, client.optimizer,
                                    _buf(client.work, 'agg',
                                         client.arena.flat))
                client.comm.flush_comm_time()
                log('Total number of samples seen on device {} is {}'.format(
                    args.graph.rank, args.local_data_seen), args.debug)
                if args.graph.rank == 0:
                    do_validate(args, client.model, client.optimizer,
                                client.criterion, client.metrics,
                                client.test_loader, None, data_mode='test')
                return

            logging_display_training(args, tracker)
            tracker['start_load_time'] = time.time()

        if args.reshuffle_per_epoch:
            log('reshuffle the dataset.', args.debug)
            del client.train_loader, client.test_loader
            gc.collect()
            client.load_local_dataset()
