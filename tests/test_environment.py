"""Environment-contract tests (superset of reference tests/test_environment.py):
construction, random-playout loops, and the network-match consistency oracle
(replicas synced only via diff_info/update must agree on legal actions)."""

import importlib
import random

import pytest

ENVS = [
    'tictactoe',
    'parallel_tictactoe',
    'geister',
    'hungry_geese',
]

N_EPISODES = 30


def _load(env):
    return importlib.import_module('handyrl_amd.envs.' + env)


@pytest.mark.parametrize('env', ENVS)
def test_environment_property(env):
    e = _load(env).Environment()
    assert isinstance(e.players(), list)
    str(e)


@pytest.mark.parametrize('env', ENVS)
def test_environment_local(env):
    e = _load(env).Environment()
    for _ in range(N_EPISODES):
        e.reset()
        steps = 0
        while not e.terminal():
            actions = {}
            for player in e.turns():
                actions[player] = random.choice(e.legal_actions(player))
            e.step(actions)
            e.reward()
            steps += 1
            assert steps < 1000, 'episode did not terminate'
        oc = e.outcome()
        assert set(oc.keys()) == set(e.players())


@pytest.mark.parametrize('env', ENVS)
def test_environment_network(env):
    mod = _load(env)
    e = mod.Environment()
    replicas = {p: mod.Environment() for p in e.players()}
    for _ in range(N_EPISODES):
        e.reset()
        for p, rep in replicas.items():
            rep.update(e.diff_info(p), True)
        while not e.terminal():
            actions = {}
            for player in e.turns():
                assert set(e.legal_actions(player)) == \
                    set(replicas[player].legal_actions(player))
                action = random.choice(replicas[player].legal_actions(player))
                actions[player] = replicas[player].action2str(action, player)
            actions = {p: e.str2action(a, p) for p, a in actions.items()}
            e.step(actions)
            for p, rep in replicas.items():
                rep.update(e.diff_info(p), False)
            e.reward()
        e.outcome()


@pytest.mark.parametrize('env', ENVS)
def test_observation_shapes_stable(env):
    """Observations keep shape/dtype across steps and players."""
    import numpy as np
    from handyrl_amd.util import map_r

    e = _load(env).Environment()
    e.reset()
    ref_shapes = map_r(e.observation(e.players()[0]), lambda o: (o.shape, o.dtype))
    for _ in range(20):
        if e.terminal():
            break
        for p in e.players():
            obs = e.observation(p)
            shapes = map_r(obs, lambda o: (o.shape, o.dtype))
            assert shapes == ref_shapes
            map_r(obs, lambda o: np.asarray(o))
        actions = {p: random.choice(e.legal_actions(p)) for p in e.turns()}
        e.step(actions)


def test_check_env_validator():
    """The runtime contract validator passes every bundled game
    (turn-based, simultaneous and 4-player survival alike)."""
    from handyrl_amd.environment import make_env, check_env
    for name in ['TicTacToe', 'ParallelTicTacToe', 'Geister',
                 'HungryGeese']:
        assert check_env(make_env({'env': name}))
