"""Environment behavioral parity against the REFERENCE env implementations
(read-only import): identical legal actions, observations, terminal flags
and outcomes under shared random play."""

import random
import sys

import numpy as np
import pytest

REFERENCE = '/root/reference'


def _ref_env(module, cls='Environment'):
    sys.path.insert(0, REFERENCE)
    try:
        mod = __import__('handyrl.envs.%s' % module, fromlist=[cls])
    finally:
        sys.path.remove(REFERENCE)
    return getattr(mod, cls)()


def _compare_obs(a, b):
    if isinstance(a, dict):
        assert set(a.keys()) == set(b.keys())
        for k in a:
            _compare_obs(a[k], b[k])
    else:
        np.testing.assert_allclose(a, b, rtol=1e-6, atol=1e-6)


@pytest.mark.parametrize('module,ours_mod,episodes,steps', [
    ('geister', 'geister', 4, 260),
    ('tictactoe', 'tictactoe', 12, 12),
    ('parallel_tictactoe', 'parallel_tictactoe', 12, 12),
])
def test_env_matches_reference(module, ours_mod, episodes, steps):
    import importlib
    ours_env = importlib.import_module('handyrl_amd.envs.%s' % ours_mod) \
        .Environment()
    ref_env = _ref_env(module)
    rng = random.Random(29)

    for ep in range(episodes):
        ours_env.reset()
        ref_env.reset()
        for t in range(steps):
            assert ours_env.terminal() == ref_env.terminal()
            if ours_env.terminal():
                break
            assert list(ours_env.turns()) == list(ref_env.turns())
            actions = {}
            for p in ours_env.turns():
                la_o = sorted(ours_env.legal_actions(p))
                la_r = sorted(ref_env.legal_actions(p))
                assert la_o == la_r, (module, ep, t, p)
                _compare_obs(ours_env.observation(p), ref_env.observation(p))
                # string round-trip parity too
                a = rng.choice(la_o)
                assert ours_env.action2str(a, p) == ref_env.action2str(a, p)
                actions[p] = a
            if len(actions) == 1:
                ((p, a),) = actions.items()
                ours_env.play(a, p)
                ref_env.play(a, p)
            else:
                # both engines draw the landing player from the global
                # RNG: replay the same state so the SAME action lands
                st = random.getstate()
                ours_env.step(actions)
                random.setstate(st)
                ref_env.step(actions)
        if ours_env.terminal():
            assert ours_env.outcome() == ref_env.outcome(), (module, ep)
