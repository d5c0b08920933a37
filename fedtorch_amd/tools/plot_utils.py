# -*- coding: utf-8 -*-
"""Plot helpers (parity with reference `fedtorch/tools/plot_utils.py`).
matplotlib is optional in this environment; functions degrade to no-ops
with a message when it is missing."""
import numpy as np

try:
    import matplotlib
    matplotlib.use('Agg')
    import matplotlib.pyplot as plt
    HAS_MPL = True
except Exception:  # pragma: no cover
    HAS_MPL = False


def smooth(ys, weight=0.6):
    """Exponential smoothing like TensorBoard (reference `plot_utils.py`)."""
    if len(ys) == 0:
        return ys
    smoothed = []
    last = ys[0]
    for y in ys:
        last = last * weight + (1 - weight) * y
        smoothed.append(last)
    return np.asarray(smoothed)


def plot_curves(curves, labels, xlabel='round', ylabel='top1', title='',
                out_path=None, smooth_weight=0.0):
    """curves: list of (x, y) arrays."""
    if not HAS_MPL:
        print('matplotlib not available; skipping plot %s' % title)
        return None
    fig, ax = plt.subplots(figsize=(6, 4))
    for (x, y), lab in zip(curves, labels):
        if smooth_weight > 0:
            y = smooth(np.asarray(y), smooth_weight)
        ax.plot(x, y, label=lab)
    ax.set_xlabel(xlabel)
    ax.set_ylabel(ylabel)
    ax.set_title(title)
    ax.legend()
    ax.grid(alpha=0.3)
    if out_path:
        fig.savefig(out_path, bbox_inches='tight', dpi=120)
        plt.close(fig)
        return out_path
    return fig
