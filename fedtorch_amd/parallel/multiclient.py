# -*- coding: utf-8 -*-
"""Virtual-client packing: C federated clients resident on ONE GPU rank.

The reference can only simulate many clients in its single-process
"centered" mode (`main_centered.py`).  On MI355X, 288 GB HBM3E holds
hundreds of ResNet-class replicas, so the packed mode is first-class: each
rank owns

* one nn.Module + arena for COMPUTE (clients time-share it),
* a ``[C, N]`` replica arena (one flat row per client),
* ``[C, N]`` momentum state (+ per-algorithm aux rows),
* per-client data partitions (global client id = rank*C + j).

A federated round runs each online local client's tau local steps on the
shared module (swap = one ``load_flat`` + momentum view bind), accumulates
the weighted diffs into ONE per-rank partial, and a single world all-reduce
finishes the aggregation — identical math to W*C one-client ranks
(`tests/test_multiclient.py` pins packed(2x2) == flat(4)).
"""
from copy import copy

import torch

from fedtorch_amd import ops


class VirtualGraph(object):
    """graph facade so the dataset pipeline partitions over W*C clients."""

    def __init__(self, base, gid, total):
        self.rank = gid
        self.world = list(range(total))
        self.on_cuda = base.on_cuda
        self.blocks = getattr(base, 'blocks', str(total))

    @property
    def n_nodes(self):
        return len(self.world)

    @property
    def ranks(self):
        return list(range(self.n_nodes))

    @property
    def device(self):
        return 0


class ClientPack(object):
    """The C virtual clients of one GPU rank."""

    def __init__(self, client, clients_per_rank):
        """client: an initialized `Client` (model/arena/optimizer built)."""
        self.base = client
        self.args = client.args
        self.C = clients_per_rank
        self.rank = client.args.graph.rank
        self.world = client.args.graph.n_nodes
        self.total_clients = self.world * self.C
        arena = client.arena
        n = arena.numel
        dev = arena.flat.device
        # per-client state resident in HBM
        self.replicas = torch.zeros((self.C, n), device=dev)
        self.in_mom = torch.zeros((self.C, n), device=dev) \
            if client.optimizer.param_groups[0]['in_momentum'] else None
        self.mom_init = [False] * self.C
        self.train_loaders = [None] * self.C
        self.partial = torch.zeros(n, device=dev)
        # per-client BatchNorm running stats (the reference's centered mode
        # gives every ClientCentered its own module and therefore its own
        # BN buffers; the shared compute module must swap them per client)
        self.bufs = None
        if arena.buf_flat is not None:
            self.bufs = torch.zeros((self.C, arena.buf_flat.numel()),
                                    device=dev)
            for c in range(self.C):
                self.bufs[c].copy_(arena.buf_flat)
        for c in range(self.C):
            self.replicas[c].copy_(arena.flat)

    def global_id(self, j):
        return self.rank * self.C + j

    def build_loaders(self):
        """per-virtual-client data partitions over W*C clients.

        ONE partitioner is built (by the first local client) and shared
        by the rank's other virtual clients — building one per client
        gave every client a DIFFERENT index permutation (only the
        global-client-0 call shuffles before the broadcast), so chunks
        could OVERLAP across virtual clients.  Sharing mirrors the
        centered mode (`main_centered.py:20-25`) and keeps the broadcast
        count identical on every rank (one per rank, at j == 0)."""
        from fedtorch_amd.components.dataset import define_dataset
        per_client_data = self.args.data in ('emnist', 'emnist_full',
                                             'synthetic', 'shakespeare')
        part = None
        for j in range(self.C):
            args_j = copy(self.args)
            args_j.graph = VirtualGraph(self.args.graph, self.global_id(j),
                                        self.total_clients)
            if per_client_data:
                loaders = define_dataset(args_j, shuffle=True, test=False)
            elif part is None:
                loaders, part = define_dataset(args_j, shuffle=True,
                                               test=False,
                                               return_partitioner=True)
            else:
                loaders = define_dataset(args_j, shuffle=True, test=False,
                                         Partitioner=part)
            self.train_loaders[j] = loaders[0]
        # counters derived for the last one apply to all (equal splits)
        self.args.num_batches_train_per_device_per_epoch = \
            len(self.train_loaders[0])

    def run_client(self, j, server_flat, local_step_fn):
        """Swap in client j, run its local steps, swap out.
        local_step_fn(loader) -> local_steps performed."""
        client = self.base
        client.arena.load_flat(server_flat)
        if self.bufs is not None:
            client.arena.buf_flat.copy_(self.bufs[j])
        if self.in_mom is not None:
            client.optimizer.bind_state(in_buf=self.in_mom[j],
                                        in_init=self.mom_init[j])
        steps = local_step_fn(self.train_loaders[j])
        if self.in_mom is not None:
            self.mom_init[j] = True
        self.replicas[j].copy_(client.arena.flat)
        if self.bufs is not None:
            self.bufs[j].copy_(client.arena.buf_flat)
        return steps

    def partial_buffers(self, online_local, total_online):
        """Write this rank's PRE-SCALED partial of the global BN-stat mean
        (sum over ONLINE local clients / total_online) into the compute
        module's buffer arena.  Zeroed when no local client is online, so
        a plain world all-reduce of the buffer arena yields the mean over
        ALL online clients regardless of how they spread across ranks
        (ranks with more online clients weigh proportionally more)."""
        if self.bufs is None:
            return
        buf = self.base.arena.buf_flat
        if not online_local or total_online <= 0:
            buf.zero_()
            return
        torch.sum(self.bufs[list(online_local)], dim=0, out=buf)
        buf.div_(float(total_online))

    def adopt_buffers(self):
        """After the world-level BN-stat all-reduce every client adopts
        the global mean (mirrors all one-client ranks ending the round
        with identical buffers)."""
        if self.bufs is None:
            return
        self.bufs.copy_(self.base.arena.buf_flat.unsqueeze(0)
                        .expand_as(self.bufs))

    def accumulate_partial(self, server_flat, weights):
        """partial = sum_j weights[j] * (server - replica_j) — ONE batched
        kernel over the [C, N] replica arena; weights: list of per-local-
        client floats (0 for offline)."""
        w = torch.tensor(weights, dtype=torch.float32,
                         device=server_flat.device)
        ops.multi_diff_accumulate(server_flat, self.replicas, w,
                                  self.partial)
        return self.partial
