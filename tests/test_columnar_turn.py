"""Turn-based columnar episodes (Geister) build batches bit-identical to
the reference moment-dict format, across window cuts and padding."""

import random

import numpy as np
import torch

from handyrl_amd.batch import make_batch, pack_moments
from handyrl_amd.envs.vec_geister import GeisterVecEnv, N_ACTIONS


def _args(**over):
    args = {
        'turn_based_training': True, 'observation': False, 'gamma': 0.8,
        'forward_steps': 8, 'burn_in_steps': 0, 'compress_steps': 4,
        'compress_episodes': False,
    }
    args.update(over)
    return args


def _record_trajectory(steps=20, seed=5):
    """One Geister game recorded in BOTH episode formats."""
    rng = np.random.default_rng(seed)
    vec = GeisterVecEnv(1, seed=seed)
    vec.reset_games(np.arange(1))
    cols = {k: [] for k in ('scalar', 'board', 'mask', 'turn', 'action',
                            'prob', 'value')}
    moments = []
    for _ in range(steps):
        mask = vec.legal_masks()[0]
        scalar, board = vec.observations()
        legal = np.nonzero(mask == 0.0)[0]
        a = int(rng.choice(legal))
        # both formats read prob/value from the float32 result shm rows
        prob = float(np.float32(rng.uniform(0.1, 1.0)))
        value = float(np.float32(rng.uniform(-1, 1)))
        p = int(vec.turn()[0])

        cols['scalar'].append(scalar[0].astype(np.uint8))
        cols['board'].append(board[0].astype(np.uint8))
        cols['mask'].append(mask == 0.0)
        cols['turn'].append(p)
        cols['action'].append(a)
        cols['prob'].append(prob)
        cols['value'].append(value)

        m = {key: {0: None, 1: None} for key in
             ('observation', 'selected_prob', 'action_mask', 'action',
              'value', 'reward', 'return')}
        m['observation'][p] = {'scalar': scalar[0].copy(),
                               'board': board[0].copy()}
        m['selected_prob'][p] = prob
        m['action_mask'][p] = mask.copy()
        m['action'][p] = a
        m['value'][p] = np.array([value], dtype=np.float32)
        m['turn'] = [p]
        m['reward'] = {0: -0.01, 1: -0.01}
        moments.append(m)
        vec.step(np.array([a]))
        assert not vec.over[0], 'trajectory ended early; lower steps'

    gamma = 0.8
    S = len(moments)
    for p in (0, 1):
        ret = 0.0
        for m in reversed(moments):
            ret = m['reward'][p] + gamma * ret
            m['return'][p] = ret
    acc, rets = 0.0, np.empty(S, np.float32)
    for t in range(S - 1, -1, -1):
        acc = -0.01 + gamma * acc
        rets[t] = acc
    outcome = {0: 1.0, 1: -1.0}
    job_args = {'player': [0, 1], 'model_id': {0: -1, 1: -1}}
    ep_dict = {'args': job_args, 'steps': S, 'outcome': outcome,
               'moment': pack_moments(moments, 4, compress=False)}
    ep_col = {'args': job_args, 'steps': S, 'outcome': outcome,
              'columnar': True, 'turn_based': True,
              'n_actions': N_ACTIONS, 'n_players': 2,
              'reward': np.full((S, 2), -0.01, np.float32),
              'return': np.stack([rets, rets], axis=1)}
    for k, v in cols.items():
        ep_col[k] = np.array(v)
    ep_col['action'] = ep_col['action'].astype(np.int16)
    ep_col['prob'] = ep_col['prob'].astype(np.float32)
    ep_col['value'] = ep_col['value'].astype(np.float32)
    return ep_dict, ep_col, S


def _window_dict(ep, st, ed, train_st, args):
    cs = args['compress_steps']
    return {'args': ep['args'], 'outcome': ep['outcome'],
            'moment': ep['moment'][st // cs:(ed - 1) // cs + 1],
            'base': (st // cs) * cs,
            'start': st, 'end': ed, 'train_start': train_st,
            'total': ep['steps']}


def _window_col(ep, st, ed, train_st, args):
    out = {'args': ep['args'], 'outcome': ep['outcome'], 'columnar': True,
           'turn_based': True, 'n_actions': ep['n_actions'],
           'n_players': ep['n_players'],
           'start': st, 'end': ed, 'train_start': train_st,
           'total': ep['steps']}
    for k in ('scalar', 'board', 'mask', 'turn', 'action', 'prob', 'value',
              'reward', 'return'):
        out[k] = ep[k][st:ed]
    return out


def _assert_batches_equal(a, b):
    assert set(a.keys()) == set(b.keys())
    for k in a:
        if isinstance(a[k], dict):
            for kk in a[k]:
                np.testing.assert_array_equal(
                    a[k][kk].numpy(), b[k][kk].numpy(), err_msg='%s.%s' % (k, kk))
        else:
            np.testing.assert_array_equal(a[k].numpy(), b[k].numpy(),
                                          err_msg=k)


def test_columnar_turn_batch_matches_dict_path():
    args = _args()
    ep_dict, ep_col, S = _record_trajectory(steps=20)
    random.seed(0)
    for st, ed, train_st in ((0, 8, 0), (5, 13, 5), (12, 20, 12)):
        bd = make_batch([_window_dict(ep_dict, st, ed, train_st, args)], args)
        bc = make_batch([_window_col(ep_col, st, ed, train_st, args)], args)
        _assert_batches_equal(bd, bc)


def test_columnar_turn_padding_matches_dict_path():
    args = _args(forward_steps=12)
    ep_dict, ep_col, S = _record_trajectory(steps=14)
    # short tail window -> padding branch
    bd = make_batch([_window_dict(ep_dict, 9, 14, 9, args)], args)
    bc = make_batch([_window_col(ep_col, 9, 14, 9, args)], args)
    _assert_batches_equal(bd, bc)


def test_columnar_turn_burn_in_matches_dict_path():
    """Recurrent configs: a burn-in prefix precedes the trained window."""
    args = _args(forward_steps=6, burn_in_steps=4)
    ep_dict, ep_col, S = _record_trajectory(steps=20)
    for st, ed, train_st in ((6, 16, 10),     # full burn-in prefix
                             (0, 8, 2),       # clipped prefix at episode start
                             (14, 20, 18)):   # short tail -> padding too
        bd = make_batch([_window_dict(ep_dict, st, ed, train_st, args)], args)
        bc = make_batch([_window_col(ep_col, st, ed, train_st, args)], args)
        _assert_batches_equal(bd, bc)


def test_columnar_turn_trains():
    from handyrl_amd.train import Trainer
    from handyrl_amd.envs.geister import Environment
    args = _args(batch_size=2, lambda_=0.7)
    args.update({'entropy_regularization': 0.1,
                 'entropy_regularization_decay': 0.1, 'lambda': 0.7,
                 'policy_target': 'UPGO', 'value_target': 'TD',
                 'seed': 0, 'bf16': False, 'num_batchers': 1,
                 'minimum_episodes': 1, 'maximum_episodes': 10})
    _, ep_col, S = _record_trajectory(steps=20)
    batch = make_batch([_window_col(ep_col, 0, 8, 0, args),
                        _window_col(ep_col, 4, 12, 4, args)], args)
    trainer = Trainer(args, Environment().net(), device=torch.device('cpu'))
    losses, dcnt = trainer.train_step(batch)
    assert torch.isfinite(losses['total'])
    assert dcnt > 0
