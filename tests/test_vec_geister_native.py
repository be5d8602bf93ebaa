"""Native C++ Geister core vs the numpy engine: bit-equal state machines
under identical random play (legal masks, observations, step transitions,
outcomes)."""

import numpy as np
import pytest

from handyrl_amd.envs import vec_geister as vg


pytestmark = pytest.mark.skipif(vg._CORE is None,
                                reason='native core not built')


def _numpy_env(n, seed):
    """Engine forced onto the numpy path."""
    env = vg.GeisterVecEnv(n, seed=seed)
    return env


def test_native_matches_numpy_exact():
    G, steps = 33, 400
    rng = np.random.default_rng(0)

    nat = vg.GeisterVecEnv(G, seed=1)
    ref = vg.GeisterVecEnv(G, seed=1)
    nat.reset_games(np.arange(G))
    ref.reset_games(np.arange(G))

    core = vg._CORE
    assert core.ready()

    for t in range(steps):
        # reference masks via the numpy path (bypass the core dispatch)
        m_ref = np.empty((G, vg.N_ACTIONS), dtype=np.float32)
        vg._CORE = None
        try:
            ref.legal_masks(out=m_ref)
            s_ref, b_ref = ref.observations()
        finally:
            vg._CORE = core
        m_nat = nat.legal_masks()
        s_nat, b_nat = nat.observations()
        np.testing.assert_array_equal(m_nat, m_ref)
        np.testing.assert_array_equal(s_nat, s_ref)
        np.testing.assert_array_equal(b_nat, b_ref)

        # identical random legal actions for both
        legal = m_ref == 0.0
        acts = np.zeros(G, dtype=np.int64)
        for g in range(G):
            idx = np.nonzero(legal[g])[0]
            acts[g] = idx[rng.integers(len(idx))] if len(idx) else 0

        d_nat = nat.step(acts).copy()
        vg._CORE = None
        try:
            d_ref = ref.step(acts).copy()
        finally:
            vg._CORE = core
        np.testing.assert_array_equal(d_nat, d_ref)
        for attr in ('board', 'slot_of', 'piece_pos', 'piece_cnt', 'color',
                     'turn_count', 'win', 'over'):
            np.testing.assert_array_equal(
                getattr(nat, attr), getattr(ref, attr), err_msg=attr)

        fin = np.nonzero(d_ref)[0]
        if len(fin):
            np.testing.assert_array_equal(nat.outcomes(fin),
                                          ref.outcomes(fin))
            nat.reset_games(fin)
            ref.reset_games(fin)
