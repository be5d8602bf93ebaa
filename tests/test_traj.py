"""Device-side trajectory recording (handyrl_amd/traj): the in-graph
recorder + commit_traj D2D ingest must fill the replay ring with exactly
the rows the host-recorded episode path would have produced (dead-seat
rows differ only where gather_batch masks them out)."""

import numpy as np
import pytest
import torch

from handyrl_amd.replay import DeviceReplay
from handyrl_amd.traj import TrajRecorder


def _args(fs=4):
    return {
        'turn_based_training': False, 'observation': False, 'gamma': 0.8,
        'forward_steps': fs, 'burn_in_steps': 0, 'compress_steps': 4,
        'batch_size': 4, 'minimum_episodes': 2, 'maximum_episodes': 64,
        'lambda': 0.7, 'policy_target': 'VTRACE', 'value_target': 'VTRACE',
    }


def _record_episode(traj, g, steps, rng):
    """Record one synthetic episode into trajectory row g; returns the
    columnar episode dict the host path would have produced."""
    obs_all = (rng.random((steps, 17, 7, 11)) < 0.2).astype(np.uint8)
    # alive is derived from head planes: force a known pattern
    alive = np.zeros((steps, 4), dtype=bool)
    for t in range(steps):
        for p in range(4):
            a = (p + t) % 4 != 3                     # seat dies sometimes
            alive[t, p] = a
            obs_all[t, p] = 0
            if a:
                obs_all[t, p, t % 7, p % 11] = 1     # head cell
    act = rng.integers(0, 4, (steps, 4)).astype(np.int32)
    prob = rng.random((steps, 4)).astype(np.float32)
    val = rng.standard_normal((steps, 4)).astype(np.float32)

    for t in range(steps):
        packed = np.stack([act[t].astype(np.float32), prob[t], val[t]],
                          axis=1).reshape(4, 3)
        traj.record_(torch.from_numpy(obs_all[t:t + 1]),
                     torch.from_numpy(packed.reshape(4, 3)),
                     torch.tensor([g], dtype=torch.int64),
                     torch.tensor([t], dtype=torch.int64))

    # host-path equivalent (dead seats zeroed the way actor.py records)
    return {
        'args': {'player': [0, 1, 2, 3],
                 'model_id': {p: -1 for p in range(4)}},
        'steps': steps, 'columnar': True, 'canonical_obs': True,
        'n_actions': 4,
        'outcome': {p: float(p) / 3 for p in range(4)},
        'obs': obs_all, 'alive': alive,
        'action': np.where(alive, act, 0),
        'prob': np.where(alive, prob, 0.0),
        'value': np.where(alive, val, 0.0),
    }


def test_commit_traj_matches_host_extend():
    rng = np.random.default_rng(3)
    dev = torch.device('cpu')
    traj = TrajRecorder(8, dev, max_steps=16)

    eps = [_record_episode(traj, g, steps, rng)
           for g, steps in [(0, 5), (3, 7), (5, 3)]]
    outcomes = np.array([[ep['outcome'][p] for p in range(4)] for ep in eps],
                        dtype=np.float32)

    r_traj = DeviceReplay(_args(), dev, bytes_budget=64 << 20)
    r_traj.commit_traj(traj, np.array([0, 3, 5], dtype=np.int64),
                       np.array([5, 7, 3], dtype=np.int64), outcomes)

    r_host = DeviceReplay(_args(), dev, bytes_budget=64 << 20)
    r_host.extend(eps)

    assert len(r_traj) == len(r_host) == 3
    assert list(r_traj.table) != []
    for (p0a, sa, oca), (p0b, sb, ocb) in zip(r_traj.table, r_host.table):
        assert (p0a, sa) == (p0b, sb)
        np.testing.assert_array_equal(oca, ocb)

    n = sum(ep['steps'] for ep in eps)
    # obs and alive identical everywhere
    torch.testing.assert_close(r_traj.obs[:n], r_host.obs[:n])
    assert torch.equal(r_traj.alive[:n], r_host.alive[:n])
    # act/prob/value identical on alive seats (dead seats are masked by
    # gather_batch; host stores zeros there, traj stores raw net outputs)
    m = r_host.alive[:n]
    assert torch.equal(r_traj.action[:n][m], r_host.action[:n][m])
    torch.testing.assert_close(r_traj.prob[:n][m], r_host.prob[:n][m])
    torch.testing.assert_close(r_traj.value[:n][m], r_host.value[:n][m])


def test_commit_traj_gather_matches_host_gather():
    """Full-batch equivalence through gather_batch (the garbage dead-seat
    rows must be invisible downstream)."""
    rng = np.random.default_rng(11)
    dev = torch.device('cpu')
    traj = TrajRecorder(8, dev, max_steps=16)
    eps = [_record_episode(traj, g, steps, rng)
           for g, steps in [(1, 6), (2, 4)]]
    outcomes = np.array([[ep['outcome'][p] for p in range(4)] for ep in eps],
                        dtype=np.float32)

    r_traj = DeviceReplay(_args(), dev, bytes_budget=64 << 20)
    r_traj.commit_traj(traj, np.array([1, 2], dtype=np.int64),
                       np.array([6, 4], dtype=np.int64), outcomes)
    r_host = DeviceReplay(_args(), dev, bytes_budget=64 << 20)
    r_host.extend(eps)

    import random as pyrandom
    B = 4
    pyrandom.seed(5)
    idx_a = r_traj.sample_indices(B)
    pyrandom.seed(5)
    idx_b = r_host.sample_indices(B)
    to_t = lambda arrs: [torch.from_numpy(np.asarray(a)) for a in arrs]
    batch_a = r_traj.gather_batch(*to_t(idx_a))
    batch_b = r_host.gather_batch(*to_t(idx_b))
    for k in batch_b:
        torch.testing.assert_close(batch_a[k], batch_b[k], rtol=0, atol=0,
                                   msg=lambda m, k=k: '%s: %s' % (k, m))


def test_extend_skips_committed_stubs():
    r = DeviceReplay(_args(), torch.device('cpu'), bytes_budget=64 << 20)
    r.extend([{'committed': True, 'steps': 5,
               'outcome': {p: 0.0 for p in range(4)}}])
    assert len(r) == 0


def _record_turn_episode(traj, g, steps, rng, gamma=0.8):
    """Record one synthetic turn-based episode into trajectory row g and
    return the equivalent columnar turn-based episode dict."""
    scalar = rng.integers(0, 2, (steps, 18)).astype(np.uint8)
    board = rng.integers(0, 2, (steps, 7, 6, 6)).astype(np.uint8)
    mask = rng.random((steps, 214)) < 0.2
    mask[:, 0] = True                               # >=1 legal
    turn = (np.arange(steps) % 2).astype(np.int8)
    act = rng.integers(0, 214, steps).astype(np.int16)
    prob = rng.random(steps).astype(np.float32)
    val = rng.standard_normal(steps).astype(np.float32)

    for t in range(steps):
        packed = np.stack([act[t].astype(np.float32), prob[t], val[t],
                           np.float32(0)]).reshape(1, 4)
        traj.record_(
            torch.from_numpy(scalar[t:t + 1].astype(np.float32)),
            torch.from_numpy(board[t:t + 1].astype(np.float32)),
            torch.from_numpy(np.where(mask[t:t + 1], 0.0, 1e32)
                             .astype(np.float32)),
            torch.tensor([int(turn[t])], dtype=torch.int64),
            torch.from_numpy(packed),
            torch.tensor([g], dtype=torch.int64),
            torch.tensor([t], dtype=torch.int64))

    acc, rets = 0.0, np.empty(steps, np.float32)
    for t in range(steps - 1, -1, -1):
        acc = -0.01 + gamma * acc
        rets[t] = acc
    return {'args': {'player': [0, 1], 'model_id': {0: -1, 1: -1}},
            'steps': steps, 'columnar': True, 'turn_based': True,
            'n_actions': 214, 'n_players': 2,
            'outcome': {0: 1.0, 1: -1.0},
            'scalar': scalar, 'board': board, 'mask': mask, 'turn': turn,
            'action': act, 'prob': prob, 'value': val,
            'reward': np.full((steps, 2), -0.01, np.float32),
            'return': np.stack([rets, rets], axis=1)}


def test_turn_commit_traj_matches_host_extend():
    from handyrl_amd.replay import TurnDeviceReplay
    from handyrl_amd.traj import GeisterTrajRecorder

    rng = np.random.default_rng(7)
    dev = torch.device('cpu')
    traj = GeisterTrajRecorder(8, dev, max_steps=32)
    args = {'turn_based_training': True, 'observation': False,
            'gamma': 0.8, 'forward_steps': 8, 'burn_in_steps': 0,
            'compress_steps': 4, 'batch_size': 4, 'minimum_episodes': 2,
            'maximum_episodes': 64, 'lambda': 0.7,
            'policy_target': 'UPGO', 'value_target': 'TD'}

    eps = [_record_turn_episode(traj, g, steps, rng)
           for g, steps in [(0, 9), (2, 5), (6, 12)]]
    outcomes = np.array([[ep['outcome'][p] for p in range(2)] for ep in eps],
                        dtype=np.float32)

    r_traj = TurnDeviceReplay(args, dev, bytes_budget=64 << 20)
    r_traj.commit_traj(traj, np.array([0, 2, 6], dtype=np.int64),
                       np.array([9, 5, 12], dtype=np.int64), outcomes)
    r_host = TurnDeviceReplay(args, dev, bytes_budget=64 << 20)
    r_host.extend(eps)

    assert len(r_traj) == len(r_host) == 3
    for (p0a, sa, oca), (p0b, sb, ocb) in zip(r_traj.table, r_host.table):
        assert (p0a, sa) == (p0b, sb)
        np.testing.assert_array_equal(oca, ocb)
    n = sum(ep['steps'] for ep in eps)
    for col in ('scalar', 'board', 'mask', 'turn', 'action'):
        assert torch.equal(getattr(r_traj, col)[:n],
                           getattr(r_host, col)[:n]), col
    torch.testing.assert_close(r_traj.prob[:n], r_host.prob[:n])
    torch.testing.assert_close(r_traj.value[:n], r_host.value[:n])
    torch.testing.assert_close(r_traj.reward[:n], r_host.reward[:n])
    # closed-form discounted return vs the serial float backfill
    torch.testing.assert_close(r_traj.ret[:n], r_host.ret[:n],
                               rtol=1e-5, atol=1e-5)

    import random as pyrandom
    to_t = lambda arrs: [torch.from_numpy(np.asarray(a)) for a in arrs]
    pyrandom.seed(3)
    idx_a = r_traj.sample_indices(4)
    pyrandom.seed(3)
    idx_b = r_host.sample_indices(4)
    ba = r_traj.gather_batch(*to_t(idx_a))
    bb = r_host.gather_batch(*to_t(idx_b))
    for k in bb:
        torch.testing.assert_close(ba[k], bb[k], rtol=1e-5, atol=1e-5,
                                   msg=lambda m, k=k: '%s: %s' % (k, m))
