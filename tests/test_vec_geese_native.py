"""The C++ env core is a BIT-EQUAL state machine to the numpy engine:
every array after every step matches exactly (the RNG-consuming phases
stay in python, so both paths read the identical stream)."""

import numpy as np
import pytest

import handyrl_amd.envs.vec_geese as vg


@pytest.mark.skipif(vg._CORE is None,
                    reason='native env core not built (envs/native_build.py)')
def test_native_step_matches_numpy_exactly():
    G = 64
    a = vg.GeeseVecEnv(G, seed=5)       # native path
    b = vg.GeeseVecEnv(G, seed=5)       # forced numpy path
    rng = np.random.default_rng(3)
    attrs = ('body', 'start', 'length', 'alive', 'scores', 'last_action',
             'prev_head', 'food', 'step_count', 'over', 'body_grid')
    resets = 0
    for t in range(400):
        acts = rng.integers(0, 4, (G, 4)).astype(np.int32)
        da = a.step(acts)
        core, vg._CORE = vg._CORE, None
        try:
            db = b.step(acts)
        finally:
            vg._CORE = core
        assert np.array_equal(da, db), t
        for attr in attrs:
            assert np.array_equal(getattr(a, attr), getattr(b, attr)), \
                (t, attr)
        oa = a.observations().copy()
        core, vg._CORE = vg._CORE, None
        try:
            ob = b.observations().copy()
        finally:
            vg._CORE = core
        assert np.array_equal(oa, ob), t
        done = np.nonzero(da)[0]
        if len(done):
            a.reset_games(done)
            b.reset_games(done)
            resets += len(done)
    assert resets > 50, 'too few episode turnovers to trust the sweep'
