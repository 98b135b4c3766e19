"""Training-progress plots (PNG per log key).

Parity: /root/reference/coinstac_dinunet/vision/plotter.py:20-65
(plot_progress): raw + rolling-mean line per log_header column group,
agg backend, one PNG per plot key.
"""
import os as _os

import numpy as _np

import matplotlib
matplotlib.use('Agg')
import matplotlib.pyplot as _plt  # noqa: E402


COLORS = ['blue', 'maroon', 'magenta', 'teal', 'red',
          'blueviolet', 'brown', 'cadetblue', 'darkgreen',
          'darkorange', 'black', 'crimson']


def _rolling_mean(x, w=10):
    if len(x) < 2:
        return x
    w = max(1, min(w, len(x)))
    kernel = _np.ones(w) / w
    return _np.convolve(x, kernel, mode='valid')


def plot_progress(cache, log_dir, plot_keys=(), epoch=None):
    """Render cache[key] rows (lists of floats) as PNG curves.

    The log_header ('Loss|Accuracy,F1') groups columns into sub-plots:
    '|'-separated groups, ','-separated labels within a group.
    """
    header = cache.get('log_header', 'Loss|Accuracy,F1')
    groups = [g.split(',') for g in header.split('|')]
    for key in plot_keys:
        rows = cache.get(key, [])
        if len(rows) < 2:
            continue
        arr = _np.asarray([r for r in rows if isinstance(r, (list, tuple))],
                          dtype=_np.float64)
        if arr.ndim != 2 or arr.shape[0] < 2:
            continue
        n_groups = len(groups)
        fig, axes = _plt.subplots(1, n_groups, figsize=(5 * n_groups, 4))
        if n_groups == 1:
            axes = [axes]
        col = 0
        for gi, labels in enumerate(groups):
            ax = axes[gi]
            for label in labels:
                if col >= arr.shape[1]:
                    break
                y = arr[:, col]
                c = COLORS[col % len(COLORS)]
                ax.plot(y, alpha=0.3, color=c)
                ax.plot(_np.arange(len(y) - len(_rolling_mean(y)), len(y)),
                        _rolling_mean(y), label=label, color=c)
                col += 1
            ax.legend()
            ax.grid(True, alpha=0.3)
            ax.set_xlabel('iteration')
        fig.tight_layout()
        _os.makedirs(log_dir, exist_ok=True)
        fig.savefig(_os.path.join(log_dir, f'{key}.png'), dpi=80)
        _plt.close(fig)
