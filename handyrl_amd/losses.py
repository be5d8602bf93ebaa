"""Off-policy target algorithms: Monte-Carlo, TD(lambda), UPGO, V-Trace.

Semantics match reference handyrl/losses.py:16-81 (V-Trace per
arXiv:1802.01561).  All tensors are (B, T, P, 1); the recurrence runs
backward over T.

Dispatch: on CUDA tensors the whole backward scan runs as ONE CDNA4 HIP
kernel (handyrl_amd/ops/src/scan.hip) with (B*P) lanes scanning T serially
in registers — replacing the reference's Python-level loop of ~3*T tiny
kernel launches per call (reference losses.py:20-29, 45-60).  The eager
loops below are the CPU path and the parity oracle for tests.
"""

import torch

from . import ops


def monte_carlo(values, returns):
    return returns, returns - values


def temporal_difference(values, returns, rewards, lambda_, gamma):
    """tv_t = r_t + gamma * ((1-lambda_{t+1}) V_{t+1} + lambda_{t+1} tv_{t+1})."""
    T = values.size(1)
    tv = returns[:, -1]
    out = [tv]
    for t in range(T - 2, -1, -1):
        r = rewards[:, t] if rewards is not None else 0
        lam = lambda_[:, t + 1]
        tv = r + gamma * ((1 - lam) * values[:, t + 1] + lam * tv)
        out.append(tv)
    targets = torch.stack(out[::-1], dim=1)
    return targets, targets - values


def upgo(values, returns, rewards, lambda_, gamma):
    """TD scan with a max(V, lambda-mixture) bootstrap (AlphaStar UPGO)."""
    T = values.size(1)
    tv = returns[:, -1]
    out = [tv]
    for t in range(T - 2, -1, -1):
        v_next = values[:, t + 1]
        r = rewards[:, t] if rewards is not None else 0
        lam = lambda_[:, t + 1]
        tv = r + gamma * torch.max(v_next, (1 - lam) * v_next + lam * tv)
        out.append(tv)
    targets = torch.stack(out[::-1], dim=1)
    return targets, targets - values


def vtrace(values, returns, rewards, lambda_, gamma, rhos, cs):
    """IMPALA V-Trace: delta scan then vs / advantage construction."""
    r = rewards if rewards is not None else 0
    v_next = torch.cat([values[:, 1:], returns[:, -1:]], dim=1)
    deltas = rhos * (r + gamma * v_next - values)

    T = values.size(1)
    acc = deltas[:, -1]
    out = [acc]
    for t in range(T - 2, -1, -1):
        acc = deltas[:, t] + gamma * lambda_[:, t + 1] * cs[:, t] * acc
        out.append(acc)
    vs_minus_v = torch.stack(out[::-1], dim=1)
    vs = vs_minus_v + values
    vs_next = torch.cat([vs[:, 1:], returns[:, -1:]], dim=1)
    advantages = r + gamma * vs_next - values
    return vs, advantages


def compute_target(algorithm, values, returns, rewards, lmb, gamma, rhos, cs, masks):
    if values is None:
        # no baseline: Monte-Carlo returns serve as target and advantage
        return returns, returns

    if algorithm == 'MC':
        return monte_carlo(values, returns)

    # outside the episode mask the scan degrades to lambda = 1 pass-through
    lambda_ = lmb + (1 - lmb) * (1 - masks)

    if values.is_cuda and ops.available():
        return ops.target_scan(algorithm, values, returns, rewards, lambda_, gamma, rhos, cs)

    if algorithm == 'TD':
        return temporal_difference(values, returns, rewards, lambda_, gamma)
    if algorithm == 'UPGO':
        return upgo(values, returns, rewards, lambda_, gamma)
    if algorithm == 'VTRACE':
        return vtrace(values, returns, rewards, lambda_, gamma, rhos, cs)
    raise ValueError('unknown target algorithm %r' % (algorithm,))
