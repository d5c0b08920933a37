# -*- coding: utf-8 -*-
"""Epoch / sync bookkeeping (parity with reference
`comms/utils/flow_utils.py:9-44`)."""


def get_current_epoch(args):
    if args.growing_batch_size:
        args.epoch_ = args.local_data_seen / args.num_samples_per_epoch
    else:
        args.epoch_ = args.local_index / \
            args.num_batches_train_per_device_per_epoch
    args.epoch = int(args.epoch_)


def get_current_local_step(args):
    try:
        return args.local_steps[args.epoch]
    except IndexError:
        return args.local_steps[-1]


def is_stop(args):
    if args.stop_criteria == 'epoch':
        return args.epoch >= args.num_epochs
    if args.stop_criteria == 'iteration':
        return args.local_index >= args.num_iterations_per_worker
    raise NotImplementedError(args.stop_criteria)


def is_sync_fed(args):
    if args.federated_sync_type == 'local_step':
        local_step = get_current_local_step(args)
        return args.local_index % local_step == 0
    if args.federated_sync_type == 'epoch':
        return args.epoch_ % args.num_epochs_per_comm == 0
    raise NotImplementedError(args.federated_sync_type)


def update_client_epoch(args):
    args.client_epoch_total += args.local_index / \
        args.num_batches_train_per_device_per_epoch
