# -*- coding: utf-8 -*-
"""Training-sanity trackers (parity with reference `fedtorch/logs/check_training.py`).

Arena-native: norms and cosines are computed on the flat arena in one
reduction each instead of per-parameter Python loops
(reference `check_training.py:22-76`).
"""
import torch

from fedtorch_amd.logs.logging import log


def check_model_at_sync(args, arena, tag=''):
    """Log param/grad norms at a sync point (reference `:22-37`)."""
    pn = arena.flat.norm().item()
    gn = arena.grad.norm().item() if arena.grad is not None else float('nan')
    log('sync-check {} rank {}: |w|={:.6f} |g|={:.6f}'.format(
        tag, args.graph.rank, pn, gn), debug=args.debug)
    return pn, gn


def track_model_aggregation(args, local_diff_flat, agg_diff_flat, init_flat,
                            current_flat):
    """Cosine(local model diff, aggregated diff) + distance from init
    (reference `:43-76`)."""
    cos = torch.nn.functional.cosine_similarity(
        local_diff_flat.view(1, -1), agg_diff_flat.view(1, -1)).item()
    dist_init = (current_flat - init_flat).norm().item()
    log('aggregation-track rank {}: cos(local, agg)={:.4f} |w - w0|={:.4f}'
        .format(args.graph.rank, cos, dist_init), debug=args.debug)
    return cos, dist_init
