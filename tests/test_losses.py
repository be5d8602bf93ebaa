"""Golden tests for the target algorithms: the vectorized/fused scans must
match a naive per-element Python reference of the published recurrences
(IMPALA V-Trace arXiv:1802.01561; TD(lambda); AlphaStar UPGO)."""

import numpy as np
import pytest
import torch

from handyrl_amd import losses


def naive_td(values, returns, rewards, lambda_, gamma):
    B, T, P, _ = values.shape
    tgt = np.zeros_like(values)
    tgt[:, -1] = returns[:, -1]
    for b in range(B):
        for p in range(P):
            for t in range(T - 2, -1, -1):
                r = rewards[b, t, p, 0] if rewards is not None else 0.0
                lam = lambda_[b, t + 1, p, 0]
                tgt[b, t, p, 0] = r + gamma * (
                    (1 - lam) * values[b, t + 1, p, 0] + lam * tgt[b, t + 1, p, 0])
    return tgt, tgt - values


def naive_upgo(values, returns, rewards, lambda_, gamma):
    B, T, P, _ = values.shape
    tgt = np.zeros_like(values)
    tgt[:, -1] = returns[:, -1]
    for b in range(B):
        for p in range(P):
            for t in range(T - 2, -1, -1):
                r = rewards[b, t, p, 0] if rewards is not None else 0.0
                lam = lambda_[b, t + 1, p, 0]
                v1 = values[b, t + 1, p, 0]
                boot = max(v1, (1 - lam) * v1 + lam * tgt[b, t + 1, p, 0])
                tgt[b, t, p, 0] = r + gamma * boot
    return tgt, tgt - values


def naive_vtrace(values, returns, rewards, lambda_, gamma, rhos, cs):
    B, T, P, _ = values.shape
    vs = np.zeros_like(values)
    adv = np.zeros_like(values)
    for b in range(B):
        for p in range(P):
            deltas = np.zeros(T)
            for t in range(T):
                r = rewards[b, t, p, 0] if rewards is not None else 0.0
                v1 = values[b, t + 1, p, 0] if t < T - 1 else returns[b, -1, p, 0]
                deltas[t] = rhos[b, t, p, 0] * (r + gamma * v1 - values[b, t, p, 0])
            vmv = np.zeros(T)
            vmv[-1] = deltas[-1]
            for t in range(T - 2, -1, -1):
                vmv[t] = deltas[t] + gamma * lambda_[b, t + 1, p, 0] * cs[b, t, p, 0] * vmv[t + 1]
            for t in range(T):
                vs[b, t, p, 0] = vmv[t] + values[b, t, p, 0]
            for t in range(T):
                r = rewards[b, t, p, 0] if rewards is not None else 0.0
                vs1 = vs[b, t + 1, p, 0] if t < T - 1 else returns[b, -1, p, 0]
                adv[b, t, p, 0] = r + gamma * vs1 - values[b, t, p, 0]
    return vs, adv


def _rand_inputs(B=5, T=9, P=2, seed=0, with_rewards=True):
    rng = np.random.default_rng(seed)
    values = rng.standard_normal((B, T, P, 1)).astype(np.float32)
    returns = rng.standard_normal((B, T, P, 1)).astype(np.float32)
    rewards = rng.standard_normal((B, T, P, 1)).astype(np.float32) if with_rewards else None
    masks = (rng.random((B, T, P, 1)) > 0.3).astype(np.float32)
    rhos = rng.random((B, T, P, 1)).astype(np.float32)
    cs = rng.random((B, T, P, 1)).astype(np.float32)
    return values, returns, rewards, masks, rhos, cs


@pytest.mark.parametrize('with_rewards', [True, False])
@pytest.mark.parametrize('algo', ['TD', 'UPGO', 'VTRACE', 'MC'])
def test_compute_target_matches_naive(algo, with_rewards):
    values, returns, rewards, masks, rhos, cs = _rand_inputs(with_rewards=with_rewards)
    lmb, gamma = 0.7, 0.9
    lambda_ = lmb + (1 - lmb) * (1 - masks)

    t = lambda a: torch.from_numpy(a) if a is not None else None
    tgt, adv = losses.compute_target(
        algo, t(values), t(returns), t(rewards), lmb, gamma, t(rhos), t(cs), t(masks))

    if algo == 'MC':
        exp_t, exp_a = returns, returns - values
    elif algo == 'TD':
        exp_t, exp_a = naive_td(values, returns, rewards, lambda_, gamma)
    elif algo == 'UPGO':
        exp_t, exp_a = naive_upgo(values, returns, rewards, lambda_, gamma)
    else:
        exp_t, exp_a = naive_vtrace(values, returns, rewards, lambda_, gamma, rhos, cs)

    np.testing.assert_allclose(tgt.numpy(), exp_t, rtol=1e-5, atol=1e-5)
    np.testing.assert_allclose(adv.numpy(), exp_a, rtol=1e-5, atol=1e-5)


def test_compute_target_no_baseline():
    values, returns, rewards, masks, rhos, cs = _rand_inputs()
    t = lambda a: torch.from_numpy(a)
    tgt, adv = losses.compute_target(
        'VTRACE', None, t(returns), t(rewards), 0.7, 0.9, t(rhos), t(cs), t(masks))
    np.testing.assert_allclose(tgt.numpy(), returns)
    np.testing.assert_allclose(adv.numpy(), returns)


@pytest.mark.gpu
@pytest.mark.parametrize('algo', ['TD', 'UPGO', 'VTRACE'])
def test_fused_scan_matches_cpu(algo):
    """HIP fused scan vs the fp32 CPU eager path on the same inputs."""
    values, returns, rewards, masks, rhos, cs = _rand_inputs(B=16, T=32, P=4, seed=3)
    lmb, gamma = 0.7, 0.9
    t = lambda a: torch.from_numpy(a) if a is not None else None
    g = lambda a: t(a).cuda() if a is not None else None

    tgt_cpu, adv_cpu = losses.compute_target(
        algo, t(values), t(returns), t(rewards), lmb, gamma, t(rhos), t(cs), t(masks))
    tgt_gpu, adv_gpu = losses.compute_target(
        algo, g(values), g(returns), g(rewards), lmb, gamma, g(rhos), g(cs), g(masks))
    torch.cuda.synchronize()

    np.testing.assert_allclose(tgt_gpu.cpu().numpy(), tgt_cpu.numpy(), rtol=1e-5, atol=1e-5)
    np.testing.assert_allclose(adv_gpu.cpu().numpy(), adv_cpu.numpy(), rtol=1e-5, atol=1e-5)
