"""Plot evaluation win rates (total and per opponent) from a learner log.

Usage: python scripts/win_rate_plot.py <train_log.txt> [out.png] [smooth_window]
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
from plot_common import parse_log, smooth


def main():
    log = sys.argv[1]
    out = sys.argv[2] if len(sys.argv) > 2 else 'win_rate_plot.png'
    window = int(sys.argv[3]) if len(sys.argv) > 3 else 5

    data = parse_log(log)
    if not data['win_rates']:
        print('no win-rate lines found in %s' % log)
        return

    import matplotlib
    matplotlib.use('Agg')
    import matplotlib.pyplot as plt

    by_opp = {}
    for epoch, opp, wr, n in data['win_rates']:
        by_opp.setdefault(opp, []).append(wr)

    fig, ax = plt.subplots(figsize=(9, 5))
    for opp, ys in sorted(by_opp.items()):
        ax.plot(smooth(ys, window), label=opp)
    ax.axhline(0.5, color='gray', lw=0.8, ls='--')
    ax.set_xlabel('epoch')
    ax.set_ylabel('win rate')
    ax.set_ylim(0, 1)
    ax.legend()
    ax.grid(alpha=0.3)
    fig.tight_layout()
    fig.savefig(out, dpi=120)
    print('wrote %s' % out)


if __name__ == '__main__':
    main()
