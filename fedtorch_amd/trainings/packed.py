# -*- coding: utf-8 -*-
"""Packed federated training: C virtual clients per GPU rank
(--clients_per_rank; BASELINE config 5 runs DRFA with 128 virtual clients
on 8 ranks).  Aggregation math is identical to W*C one-client ranks: each
rank reduces its local clients' weighted diffs into ONE partial and a
single world all-reduce finishes the sum.
"""
import time

import torch

from fedtorch_amd import ops
from fedtorch_amd.components.scheduler import adjust_learning_rate
from fedtorch_amd.components.dataset import load_data_batch
from fedtorch_amd.trainings.flow import (
    get_current_epoch, is_sync_fed)
from fedtorch_amd.trainings.eval import inference
from fedtorch_amd.trainings.federated import amp
from fedtorch_amd.trainings.afl import lambda_dual_update
from fedtorch_amd.aggregation.federated import distribute_model_server
from fedtorch_amd.logs.logging import log, logging_load_time
from fedtorch_amd.logs.meter import define_local_training_tracker


def _local_steps_fn(client, tracker):
    """Returns a closure running tau local steps on a given loader."""
    args = client.args

    def run(loader):
        steps = 0
        is_sync = False
        while not is_sync:
            for _input, _target in loader:
                steps += 1
                client.model.train()
                logging_load_time(tracker)
                args.local_index += 1
                args.local_data_seen += len(_target)
                get_current_epoch(args)
                adjust_learning_rate(args, client.optimizer,
                                     client.scheduler)
                _input, _target = load_data_batch(args, _input, _target,
                                                  tracker)
                if _input.size(0) == 1:
                    is_sync = is_sync_fed(args)
                    break
                client.optimizer.zero_grad()
                with amp(args):
                    loss, _ = inference(client.model, client.criterion,
                                        client.metrics, _input, _target,
                                        rnn=args.arch == 'rnn')
                loss.backward()
                client.optimizer.step(
                    apply_lr=True, apply_in_momentum=args.in_momentum,
                    apply_out_momentum=False)
                tracker['start_load_time'] = time.time()
                is_sync = is_sync_fed(args)
                if is_sync:
                    break
        return steps
    return run


def _apply_aggregate(client, server_flat, agg):
    """server -= lr_scale_at_sync * agg, with out momentum on the server
    state (one fused kernel)."""
    args = client.args
    g = client.optimizer.param_groups[0]
    out_m = g['out_momentum'] if args.out_momentum else 0.0
    if out_m != 0.0 and 'srv_out_mom' not in client.work:
        client.work['srv_out_mom'] = torch.zeros_like(server_flat)
        client.work['srv_out_init'] = False
    ops.fused_sgd_step(
        server_flat, agg, lr=0.0, scale=args.lr_scale_at_sync,
        weight_decay=0.0, in_momentum=0.0, out_momentum=out_m,
        dampening=0.0, nesterov=False, apply_lr=False,
        apply_in_momentum=False, apply_out_momentum=out_m != 0.0,
        out_buf=client.work.get('srv_out_mom'),
        first_out=out_m != 0.0 and not client.work.get('srv_out_init', True))
    if out_m != 0.0:
        client.work['srv_out_init'] = True


def train_and_validate_federated_packed(client, pack, validate=False):
    """FedAvg (optionally DRFA lambda-weighted) over packed virtual
    clients."""
    args = client.args
    drfa = args.federated_drfa
    total = pack.total_clients
    log('packed federated training: {} ranks x {} virtual clients'.format(
        args.graph.n_nodes, pack.C), args.debug)
    tracker = define_local_training_tracker()
    tracker['start_load_time'] = time.time()
    server = client.model_server

    if drfa:
        lambda_vector = torch.full((total,), 1.0 / total)
        kth = torch.zeros_like(pack.replicas)
        kth_avg = torch.zeros_like(server)

    step_fn = _local_steps_fn(client, tracker)
    for n_c in range(args.num_comms):
        args.rounds_comm += 1
        args.comm_time.append(0.0)
        online = client.comm.sample_online_global(total)
        distribute_model_server(client.comm, server)
        if drfa:
            client.comm.broadcast(lambda_vector, src=0)
            k_cut = torch.randint(low=1, high=max(args.local_step, 2),
                                  size=(1,))
            client.comm.broadcast(k_cut, src=0)
            k_cut = int(k_cut[0])

        # weight denominator: DISTRIBUTED semantics — global client 0
        # (the server) counts even when not sampled (`fedavg.py:17-27`,
        # same rule as aggregation/federated.rank_weight)
        n_online = len(online) if 0 in online else len(online) + 1
        n_online = max(n_online, 1)
        weights = []
        for j in range(pack.C):
            gid = pack.global_id(j)
            if gid in online:
                if drfa:
                    w = float(lambda_vector[gid]) * total / n_online

                    def step_and_snapshot(loader, j=j):
                        steps = 0
                        is_sync = False
                        while not is_sync:
                            for _input, _target in loader:
                                steps += 1
                                if steps == k_cut:
                                    kth[j].copy_(client.arena.flat)
                                client.model.train()
                                args.local_index += 1
                                args.local_data_seen += len(_target)
                                get_current_epoch(args)
                                adjust_learning_rate(
                                    args, client.optimizer, client.scheduler)
                                _input, _target = load_data_batch(
                                    args, _input, _target, tracker)
                                if _input.size(0) == 1:
                                    is_sync = is_sync_fed(args)
                                    break
                                client.optimizer.zero_grad()
                                with amp(args):
                                    loss, _ = inference(
                                        client.model, client.criterion,
                                        client.metrics, _input, _target)
                                loss.backward()
                                client.optimizer.step(
                                    apply_lr=True,
                                    apply_in_momentum=args.in_momentum,
                                    apply_out_momentum=False)
                                is_sync = is_sync_fed(args)
                                if is_sync:
                                    break
                        return steps
                    pack.run_client(j, server, step_and_snapshot)
                else:
                    w = 1.0 / n_online
                    pack.run_client(j, server, step_fn)
            else:
                w = 0.0
            weights.append(w)

        partial = pack.accumulate_partial(server, weights)
        client.comm.all_reduce(partial)
        _apply_aggregate(client, server, partial)
        client.arena.load_flat(server)
        # BN running stats: each rank contributes its pre-scaled partial
        # (sum over its online clients / total online) so the plain world
        # all-reduce gives the mean over ALL online clients even when
        # ranks have unequal online counts; then every client adopts the
        # mean (same end state as W*C one-client ranks)
        online_local = [j for j in range(pack.C)
                        if pack.global_id(j) in online]
        pack.partial_buffers(online_local, len(online))
        from fedtorch_amd.aggregation.federated import aggregate_bn_buffers
        aggregate_bn_buffers(args, client.comm, client.arena,
                             list(range(args.graph.n_nodes)),
                             work=client.work, prescaled=True)
        pack.adopt_buffers()
        for j in range(pack.C):
            # every client re-syncs at round end (its next round reloads
            # the server anyway; keeps replicas bounded in staleness)
            if weights[j] != 0.0:
                pack.replicas[j].copy_(server)

        if drfa:
            # kth average over online clients
            kth_avg.zero_()
            for j in range(pack.C):
                if weights[j] != 0.0:
                    kth_avg.add_(kth[j], alpha=1.0 / n_online)
            client.comm.all_reduce(kth_avg)
            # lambda update on a second sampled set
            online_l = client.comm.sample_online_global(total)
            loss_vec = torch.zeros(total)
            saved = client.arena.clone_flat()
            client.arena.load_flat(kth_avg)
            for j in range(pack.C):
                gid = pack.global_id(j)
                if gid not in online_l:
                    continue
                for _input, _target in pack.train_loaders[j]:
                    _input, _target = load_data_batch(args, _input, _target,
                                                      tracker)
                    if _input.size(0) == 1:
                        break
                    client.model.eval()
                    with torch.no_grad(), amp(args):
                        loss, _ = inference(client.model, client.criterion,
                                            client.metrics, _input, _target)
                    client.model.train()
                    loss_vec[gid] = loss.item() * (total / len(online_l))
                    break
            client.arena.load_flat(saved)
            client.comm.all_reduce(loss_vec)
            if args.graph.rank == 0:
                lambda_vector = lambda_dual_update(
                    args, lambda_vector, loss_vec,
                    step_scale=args.local_step)
        client.comm.flush_comm_time()
        log('packed round {} done (online {} of {}).'.format(
            n_c + 1, len(online), total), args.debug)
