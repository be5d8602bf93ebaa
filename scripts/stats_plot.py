"""Plot self-play generation outcome stats (mean +- std) from a learner log.

Usage: python scripts/stats_plot.py <train_log.txt> [out.png] [smooth_window]
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
from plot_common import parse_log, smooth


def main():
    log = sys.argv[1]
    out = sys.argv[2] if len(sys.argv) > 2 else 'stats_plot.png'
    window = int(sys.argv[3]) if len(sys.argv) > 3 else 5

    data = parse_log(log)
    if not data['gen_stats']:
        print('no generation-stats lines found in %s' % log)
        return

    import matplotlib
    matplotlib.use('Agg')
    import matplotlib.pyplot as plt

    means = smooth([m for _, m, _ in data['gen_stats']], window)
    stds = smooth([s for _, _, s in data['gen_stats']], window)

    fig, ax = plt.subplots(figsize=(9, 5))
    xs = range(len(means))
    ax.plot(xs, means, label='generation outcome mean')
    ax.fill_between(xs, [m - s for m, s in zip(means, stds)],
                    [m + s for m, s in zip(means, stds)], alpha=0.2)
    ax.set_xlabel('epoch')
    ax.set_ylabel('outcome')
    ax.legend()
    ax.grid(alpha=0.3)
    fig.tight_layout()
    fig.savefig(out, dpi=120)
    print('wrote %s' % out)


if __name__ == '__main__':
    main()
