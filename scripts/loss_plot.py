"""Plot per-term training losses from a learner stdout log.

Usage: python scripts/loss_plot.py <train_log.txt> [out.png] [smooth_window]
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
from plot_common import parse_log, smooth


def main():
    log = sys.argv[1]
    out = sys.argv[2] if len(sys.argv) > 2 else 'loss_plot.png'
    window = int(sys.argv[3]) if len(sys.argv) > 3 else 5

    data = parse_log(log)
    if not data['losses']:
        print('no loss lines found in %s' % log)
        return

    import matplotlib
    matplotlib.use('Agg')
    import matplotlib.pyplot as plt

    keys = sorted({k for _, kv in data['losses'] for k in kv})
    fig, ax = plt.subplots(figsize=(9, 5))
    for key in keys:
        ys = [kv.get(key) for _, kv in data['losses'] if key in kv]
        ax.plot(smooth(ys, window), label=key)
    ax.set_xlabel('epoch')
    ax.set_ylabel('loss (per-sample)')
    ax.legend()
    ax.grid(alpha=0.3)
    fig.tight_layout()
    fig.savefig(out, dpi=120)
    print('wrote %s' % out)


if __name__ == '__main__':
    main()
