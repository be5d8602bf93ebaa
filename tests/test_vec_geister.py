"""Parity of the vectorized Geister engine against the single-game oracle
(handyrl_amd.envs.geister.Environment): legal masks, observations, win
detection and outcomes under shared random play."""

import random

import numpy as np

from handyrl_amd.envs import geister as gz
from handyrl_amd.envs.vec_geister import (GeisterVecEnv, N_ACTIONS,
                                          N_MOVE_ACTIONS)


def oracle_mask(env):
    mask = np.full(N_ACTIONS, 1e32, dtype=np.float32)
    mask[env.legal_actions()] = 0.0
    return mask


def oracle_obs(env):
    obs = env.observation(env.turn())
    return obs['scalar'], obs['board']


def test_vec_matches_single_game_engine():
    G = 16
    rng = random.Random(11)
    vec = GeisterVecEnv(G, seed=11)
    vec.reset_games(np.arange(G))
    envs = [gz.Environment() for _ in range(G)]
    finished = 0

    for step in range(450):
        masks = vec.legal_masks()
        scalar, board = vec.observations()
        actions = np.zeros(G, dtype=np.int64)
        for g, env in enumerate(envs):
            legal = env.legal_actions()
            vec_legal = np.nonzero(masks[g] == 0.0)[0]
            assert sorted(legal) == vec_legal.tolist(), \
                'legal mismatch g=%d step=%d' % (g, step)
            o_scalar, o_board = oracle_obs(env)
            np.testing.assert_array_equal(scalar[g], o_scalar)
            np.testing.assert_array_equal(board[g], o_board)
            assert int(vec.turn()[g]) == env.turn()
            actions[g] = rng.choice(legal)
            env.play(actions[g])
        done = vec.step(actions)
        for g, env in enumerate(envs):
            assert bool(done[g]) == env.terminal(), \
                'terminal mismatch g=%d step=%d' % (g, step)
            if done[g]:
                oc = vec.outcomes(np.array([g]))[0]
                ref = env.outcome()
                assert oc[0] == ref[0] and oc[1] == ref[1]
                finished += 1
                env.reset()
                vec.reset_games(np.array([g]))
    assert finished >= 4, 'too few finished games to trust the parity sweep'


def test_vec_piece_invariants():
    G = 8
    rng = random.Random(3)
    vec = GeisterVecEnv(G, seed=3)
    vec.reset_games(np.arange(G))
    for step in range(300):
        masks = vec.legal_masks()
        actions = np.array([rng.choice(np.nonzero(masks[g] == 0.0)[0])
                            for g in range(G)], dtype=np.int64)
        vec.step(actions)
        done_idx = np.nonzero(vec.over)[0]
        started = vec.turn_count >= 0
        # counts match the board contents for every running game
        for g in np.nonzero(started & ~vec.over)[0]:
            b = vec.board[g]
            for code in range(4):
                assert int(vec.piece_cnt[g, code]) == int((b == code).sum())
            # slot table round-trips
            for cell in np.nonzero(b >= 0)[0]:
                slot = int(vec.slot_of[g, cell])
                assert int(vec.piece_pos[g, slot]) == cell
        if len(done_idx):
            vec.reset_games(done_idx)


def test_layout_turns_then_moves():
    vec = GeisterVecEnv(2, seed=0)
    vec.reset_games(np.arange(2))
    m = vec.legal_masks()
    assert (m[:, :N_MOVE_ACTIONS] == 1e32).all()
    assert (m[:, N_MOVE_ACTIONS:] == 0.0).all()
    vec.step(np.array([N_MOVE_ACTIONS, N_MOVE_ACTIONS + 69]))
    m = vec.legal_masks()
    assert (m[:, :N_MOVE_ACTIONS] == 1e32).all()     # second layout turn
    vec.step(np.array([N_MOVE_ACTIONS + 5, N_MOVE_ACTIONS + 17]))
    m = vec.legal_masks()
    assert (m[:, N_MOVE_ACTIONS:] == 1e32).all()
    assert (m[:, :N_MOVE_ACTIONS] == 0.0).any(axis=1).all()
    assert (vec.piece_cnt[:, :] == 4).all()


def _paired_after_layouts(layout0=0, layout1=0):
    """A vec game and an oracle env advanced through identical layouts."""
    vec = GeisterVecEnv(1, seed=0)
    vec.reset_games(np.arange(1))
    env = gz.Environment()
    for a in (N_MOVE_ACTIONS + layout0, N_MOVE_ACTIONS + layout1):
        env.play(a)
        vec.step(np.array([a]))
    return vec, env


def _teleport(vec, env, fx, fy, tx, ty):
    """Move a piece in BOTH engines (white-box test setup)."""
    env._relocate(fx, fy, tx, ty)
    f, t = fx * 6 + fy, tx * 6 + ty
    code, slot = vec.board[0, f], vec.slot_of[0, f]
    vec.board[0, f] = -1
    vec.slot_of[0, f] = -1
    vec.board[0, t] = code
    vec.slot_of[0, t] = slot
    vec.piece_pos[0, slot] = t


def _remove(vec, env, x, y):
    env._remove(x, y)
    c = x * 6 + y
    code, slot = int(vec.board[0, c]), int(vec.slot_of[0, c])
    vec.board[0, c] = -1
    vec.slot_of[0, c] = -1
    vec.piece_pos[0, slot] = -1
    vec.piece_cnt[0, code] -= 1


def test_blue_goal_exit_wins():
    vec, env = _paired_after_layouts()
    # layout 0 puts BLACK blues on B2,C2,D2,E2; walk the B2 blue to A6
    _teleport(vec, env, 1, 1, 0, 5)          # B2 -> A6 (goal-adjacent)
    a = env._compose_action(0, 5, 0, gz.BLACK)   # step off through (-1, 5)
    assert env.legal(a)
    assert vec.legal_masks()[0, a] == 0.0
    env.play(a)
    vec.step(np.array([a]))
    assert env.terminal() and bool(vec.over[0])
    assert env.outcome() == {0: 1, 1: -1}
    oc = vec.outcomes(np.array([0]))[0]
    assert oc[0] == 1.0 and oc[1] == -1.0


def test_capturing_all_reds_loses():
    vec, env = _paired_after_layouts()
    # layout 0: WHITE reds sit on the last four start squares (E6,D6,C6,B6)
    for sq in ('E6', 'D6', 'C6'):
        x, y = env._sq_parse(sq)
        _remove(vec, env, x, y)
    # move a BLACK piece next to white's last red at B6 (x=1, y=5)
    _teleport(vec, env, 1, 1, 1, 4)          # B2 -> B5
    a = env._compose_action(1, 4, 2, gz.BLACK)   # B5 -> B6 capture
    assert env.legal(a)
    assert vec.legal_masks()[0, a] == 0.0
    env.play(a)
    vec.step(np.array([a]))
    assert env.terminal() and bool(vec.over[0])
    # black captured ALL of white's reds -> black LOSES
    assert env.outcome() == {0: -1, 1: 1}
    oc = vec.outcomes(np.array([0]))[0]
    assert oc[0] == -1.0 and oc[1] == 1.0


def test_capturing_all_blues_wins():
    vec, env = _paired_after_layouts()
    # layout 0: WHITE blues sit on E5,D5,C5,B5
    for sq in ('E5', 'D5', 'C5'):
        x, y = env._sq_parse(sq)
        _remove(vec, env, x, y)
    # move a BLACK piece next to white's last blue at B5 (x=1, y=4)
    _teleport(vec, env, 1, 1, 1, 3)          # B2 -> B4
    a = env._compose_action(1, 3, 2, gz.BLACK)   # B4 -> B5 capture
    assert env.legal(a)
    assert vec.legal_masks()[0, a] == 0.0
    env.play(a)
    vec.step(np.array([a]))
    assert env.terminal() and bool(vec.over[0])
    # black captured ALL of white's blues -> black WINS
    assert env.outcome() == {0: 1, 1: -1}
    oc = vec.outcomes(np.array([0]))[0]
    assert oc[0] == 1.0 and oc[1] == -1.0
