# -*- coding: utf-8 -*-
"""Logical process topology (parity with reference `fedtorch/utils/topology.py`).

The reference models a fully-connected graph over MPI ranks with an optional
rank->block partition (`topology.py:57-111`).  Here the physical mapping is
one process per MI355X GPU (LOCAL_RANK -> device), with any number of virtual
clients packed per rank.
"""
import os
import functools


class FCGraph(object):
    """Fully-connected logical graph over world ranks."""

    def __init__(self, rank, blocks, on_cuda, world=None, n_nodes=None):
        self.rank = rank
        self.on_cuda = on_cuda
        self.blocks = blocks
        if world is not None and world != '':
            self.world = [int(x) for x in str(world).split(',')]
        else:
            n = n_nodes if n_nodes is not None else _world_size()
            self.world = list(range(n))

    @property
    def n_nodes(self):
        return len(self.world)

    @property
    def ranks(self):
        return list(range(self.n_nodes))

    @functools.cached_property
    def ranks_with_blocks(self):
        """rank -> block id, from the '--blocks a,b,...' string."""
        blocks = [int(b) for b in str(self.blocks).split(',')]
        # pad/truncate so that sum(blocks) covers the world.
        mapping = {}
        rank = 0
        bid = 0
        while rank < self.n_nodes:
            size = blocks[bid % len(blocks)]
            for _ in range(size):
                if rank >= self.n_nodes:
                    break
                mapping[rank] = bid
                rank += 1
            bid += 1
        return mapping

    @functools.cached_property
    def blocks_with_ranks(self):
        inv = {}
        for r, b in self.ranks_with_blocks.items():
            inv.setdefault(b, []).append(r)
        return inv

    @property
    def device(self):
        """GPU index for this rank: LOCAL_RANK if launched via torchrun,
        else rank modulo visible device count."""
        local = os.environ.get('LOCAL_RANK')
        if local is not None:
            return int(local)
        try:
            import torch
            n = torch.cuda.device_count()
        except Exception:
            n = 0
        return self.rank % n if n > 0 else self.rank

    @property
    def block(self):
        return self.ranks_with_blocks.get(self.rank, 0)

    def get_neighborhood(self):
        """Fully connected: everyone is a neighbor."""
        return [r for r in self.ranks if r != self.rank]

    def __repr__(self):
        return 'FCGraph(rank={}, n_nodes={}, device={})'.format(
            self.rank, self.n_nodes, self.device)


def _world_size():
    import torch.distributed as dist
    if dist.is_available() and dist.is_initialized():
        return dist.get_world_size()
    return int(os.environ.get('WORLD_SIZE', '1'))


def define_graph_topology(rank, blocks, on_cuda, world=None, n_nodes=None):
    return FCGraph(rank, blocks, on_cuda, world, n_nodes)
