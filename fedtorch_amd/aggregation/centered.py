# -*- coding: utf-8 -*-
"""Centered (single-process simulation) aggregation on flat arenas
(parity with reference `comms/algorithms/federated/centered/*.py`).

Same math as the reference, without `dist`: the server accumulates weighted
client diffs into a flat gradient buffer, then applies its dual-mode
optimizer step.  NOTE the reference's centered error-feedback formulas
intentionally differ from its distributed ones (centered `fedgate.py:47-49`
has no 1/w scaling); both are preserved per-mode.
"""
import numpy as np
import torch

from fedtorch_amd import ops


def set_online_clients_centered(args):
    ranks_shuffled = np.random.permutation(args.graph.ranks)
    online = ranks_shuffled[:int(args.online_client_rate *
                                 len(args.graph.ranks))]
    return sorted(online.tolist())


def _n_online(online_clients):
    """Weight denominator, DISTRIBUTED semantics (`fedavg.py:17-27`):
    client 0 doubles as the server and counts toward the denominator
    even when it is not sampled (it joins with weight 0).  The
    reference's centered mode divides by len(online) instead, so its two
    modes disagree whenever client 0 sits out a round — the oracle
    (tests/test_equivalence.py, partial participation) pins the centered
    simulator to the real distributed behavior."""
    return len(online_clients) if 0 in online_clients \
        else len(online_clients) + 1


def _weights(args, online_clients, lambda_weight):
    n_online = _n_online(online_clients)
    if lambda_weight is None:
        return {o: 1.0 / n_online for o in online_clients}
    return {o: float(lambda_weight[o]) * args.graph.n_nodes / n_online
            for o in online_clients}


def fedavg_aggregation_centered(Clients, Server, online_clients,
                                lambda_weight=None):
    """reference centered `fedavg.py:6-47` (+ fedadam, + quantized)."""
    args = Server.args
    w = _weights(args, online_clients, lambda_weight)
    agg = Server.grad
    agg.zero_()
    for o in online_clients:
        diff = Server.work.setdefault('diff', Server.arena.new_buffer())
        ops.scaled_diff(Server.arena.flat, Clients[o].arena.flat, diff, w[o])
        if args.quantized:
            q, info = ops.quantize(diff, args.quantized_bits)
            diff = ops.dequantize(q, info)
        agg.add_(diff)
    if args.federated_type == 'fedadam':
        # normalize the AGGREGATE with the server-side v state — the
        # reference applies fedadam on the summed diff at the server
        # (`federated/fedavg.py:81-85`); normalizing per-client diffs
        # (the r1 behavior) made the two execution modes disagree
        # (caught by tests/test_equivalence.py)
        from fedtorch_amd.aggregation.federated import _fedadam_normalize
        _fedadam_normalize(Server.args, Server.arena, agg)
    Server.optimizer.step(apply_lr=False, scale=args.lr_scale_at_sync,
                          apply_in_momentum=False,
                          apply_out_momentum=args.out_momentum, grad=agg)


def fedgate_aggregation_centered(Clients, Server, online_clients, local_steps,
                                 lr, lambda_weight=None):
    """reference centered `fedgate.py:7-58`."""
    args = Server.args
    n_online = _n_online(online_clients)
    w = (1.0 / n_online) if lambda_weight is None else None
    agg = Server.grad
    agg.zero_()
    diff = Server.work.setdefault('diff', Server.arena.new_buffer())
    g = Server.work.setdefault('g', Server.arena.new_buffer())
    for o in online_clients:
        wo = w if w is not None else \
            float(lambda_weight[o]) * args.graph.n_nodes / n_online
        ops.scaled_diff(Server.arena.flat, Clients[o].arena.flat, diff, wo)
        if args.quantized:
            q, info = ops.quantize(diff, args.quantized_bits)
            agg.add_(ops.dequantize(q, info))
        elif args.compressed:
            torch.add(diff, Clients[o].model_memory, alpha=wo, out=g)
            k = max(int(g.numel() * args.compressed_ratio / 2), 1)
            v, i = ops.topk_compress(g, k)
            dec = Server.work.setdefault('dec', Server.arena.new_buffer())
            ops.scatter_accumulate(dec, v.unsqueeze(0), i.unsqueeze(0))
            agg.add_(dec)
        else:
            agg.add_(diff)
    for o in online_clients:
        # delta += (server - agg - client)/(lr*tau); centered memory update
        # has NO rank-weight scaling (reference centered `fedgate.py:47-49`)
        ops.delta_update(Clients[o].model_delta, Server.arena.flat, agg,
                         Clients[o].arena.flat, 1.0 / (lr * local_steps))
        if args.compressed:
            ops.delta_update(Clients[o].model_memory, Server.arena.flat, agg,
                             Clients[o].arena.flat, 1.0)
    Server.optimizer.step(apply_lr=False, scale=args.lr_scale_at_sync,
                          apply_in_momentum=False,
                          apply_out_momentum=args.out_momentum, grad=agg)


def scaffold_aggregation_centered(Clients, Server, online_clients,
                                  local_steps, lr, lambda_weight=None):
    """reference centered `scaffold.py:3-49`."""
    args = Server.args
    n_online = _n_online(online_clients)
    agg = Server.grad
    agg.zero_()
    diff = Server.work.setdefault('diff', Server.arena.new_buffer())
    ctrl_new = Server.work.setdefault('ctrl_new', Server.arena.new_buffer())
    ctrl_delta = Server.work.setdefault('ctrl_delta',
                                        Server.arena.new_buffer())
    ctrl_delta.zero_()
    for o in online_clients:
        wo = (1.0 / n_online) if lambda_weight is None else \
            float(lambda_weight[o]) * args.graph.n_nodes / n_online
        # c+ = c - c_server + (server - client)/(tau*lr)
        ops.scaffold_control_update(ctrl_new, Clients[o].model_client_control,
                                    Server.model_server_control,
                                    Server.arena.flat, Clients[o].arena.flat,
                                    1.0 / (local_steps * lr))
        ops.scaled_diff(Server.arena.flat, Clients[o].arena.flat, diff, wo)
        agg.add_(diff)
        # accumulate (c+ - c); applied to the server control AFTER the loop
        # so every client's c+ uses the round-start server control — the
        # reference centered `scaffold.py:33` updates scp mid-loop, which
        # makes later clients see earlier clients' deltas and its own two
        # execution modes disagree; the distributed (parallel) semantics
        # is the real federated behavior (oracle: tests/test_equivalence.py)
        ctrl_delta.add_(ctrl_new).sub_(Clients[o].model_client_control)
        Clients[o].model_client_control.copy_(ctrl_new)
    # server control += sum (c+ - c) / num_workers
    ops.axpby(Server.model_server_control, ctrl_delta,
              a=1.0 / args.num_workers, b=1.0)
    Server.optimizer.step(apply_lr=False, scale=args.lr_scale_at_sync,
                          apply_in_momentum=False,
                          apply_out_momentum=args.out_momentum, grad=agg)


def qsparse_aggregation_centered(Clients, Server, online_clients, local_steps,
                                 lr, lambda_weight=None):
    """reference centered `qsparse.py:5-47`."""
    args = Server.args
    n_online = _n_online(online_clients)
    agg = Server.grad
    agg.zero_()
    diff = Server.work.setdefault('diff', Server.arena.new_buffer())
    g = Server.work.setdefault('g', Server.arena.new_buffer())
    dec = Server.work.setdefault('dec', Server.arena.new_buffer())
    for o in online_clients:
        wo = (1.0 / n_online) if lambda_weight is None else \
            float(lambda_weight[o]) * args.graph.n_nodes / n_online
        ops.scaled_diff(Server.arena.flat, Clients[o].arena.flat, diff, wo)
        torch.add(diff, Clients[o].model_memory, alpha=wo, out=g)
        k = max(int(g.numel() * args.compressed_ratio / 2), 1)
        v, i = ops.topk_compress(g, k)
        ops.scatter_accumulate(dec, v.unsqueeze(0), i.unsqueeze(0))
        agg.add_(dec)
    for o in online_clients:
        ops.delta_update(Clients[o].model_memory, Server.arena.flat, agg,
                         Clients[o].arena.flat, 1.0)
    Server.optimizer.step(apply_lr=False, scale=args.lr_scale_at_sync,
                          apply_in_momentum=False,
                          apply_out_momentum=args.out_momentum, grad=agg)


def qffl_aggregation_centered(Clients, Server, online_clients, lr):
    """reference centered `qffl.py:4-33` (arXiv:1905.10497): loss^q-weighted
    diffs with the Lipschitz normalizer h."""
    args = Server.args
    agg = Server.grad
    agg.zero_()
    diff = Server.work.setdefault('diff', Server.arena.new_buffer())
    h = 0.0
    q = args.qffl_q
    for o in online_clients:
        Fq = float(np.float_power(Clients[o].full_loss + 1e-10, q))
        ops.scaled_diff(Server.arena.flat, Clients[o].arena.flat, diff,
                        Fq / lr)
        agg.add_(diff)
        h += q * float(np.float_power(Clients[o].full_loss + 1e-10, q - 1.0)) \
            * float(diff.norm().pow(2))
        h += Fq / lr
    agg.div_(h + 1e-10)
    Server.optimizer.step(apply_lr=False, scale=args.lr_scale_at_sync,
                          apply_in_momentum=False,
                          apply_out_momentum=args.out_momentum, grad=agg)


def aggregate_kth_model_centered(Clients, Server, online_clients):
    """DRFA kth-model average (reference centered `misc.py:6-16`)."""
    w = 1.0 / len(online_clients)
    Server.kth_model.zero_()
    for o in online_clients:
        Server.kth_model.add_(Clients[o].kth_model, alpha=w)
