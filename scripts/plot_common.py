"""Shared stdout-log parsing for the plot tools.

The learner's stdout format ('updated model(N)', 'loss = k:v ...',
'win rate... = ...', 'generation stats = ...', 'epoch N') is a de-facto
public interface kept compatible with the reference so existing logs and
these tools interoperate.
"""

import re

LOSS_RE = re.compile(r'loss = (.+)')
WIN_RATE_RE = re.compile(r'win rate(?: \((.+?)\))? = ([0-9.]+) \(([-0-9.]+) / ([0-9]+)\)')
GEN_STATS_RE = re.compile(r'generation stats = ([-0-9.]+) \+- ([0-9.]+)')
EPOCH_RE = re.compile(r'epoch ([0-9]+)')
UPDATE_RE = re.compile(r'updated model\(([0-9]+)\)')


def parse_log(path):
    """Returns dict with per-epoch series parsed from a training log."""
    epochs, losses, win_rates, gen_stats, steps = [], [], [], [], []
    current_epoch = None
    with open(path) as f:
        for line in f:
            m = EPOCH_RE.search(line)
            if m:
                current_epoch = int(m.group(1))
                epochs.append(current_epoch)
            m = UPDATE_RE.search(line)
            if m:
                steps.append(int(m.group(1)))
            m = LOSS_RE.search(line)
            if m:
                kv = {}
                for part in m.group(1).split():
                    if ':' in part:
                        k, v = part.split(':', 1)
                        try:
                            kv[k] = float(v)
                        except ValueError:
                            pass
                if kv:
                    losses.append((current_epoch, kv))
            m = WIN_RATE_RE.search(line)
            if m:
                win_rates.append((current_epoch, m.group(1) or 'total',
                                  float(m.group(2)), int(m.group(4))))
            m = GEN_STATS_RE.search(line)
            if m:
                gen_stats.append((current_epoch, float(m.group(1)), float(m.group(2))))
    return {'epochs': epochs, 'losses': losses, 'win_rates': win_rates,
            'gen_stats': gen_stats, 'steps': steps}


def smooth(values, window):
    if window <= 1 or len(values) == 0:
        return list(values)
    out = []
    acc = 0.0
    from collections import deque
    q = deque()
    for v in values:
        q.append(v)
        acc += v
        if len(q) > window:
            acc -= q.popleft()
        out.append(acc / len(q))
    return out
