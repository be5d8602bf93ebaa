"""Game-environment contract and registry.

The method names and semantics of ``BaseEnvironment`` are a compatibility
contract with the reference framework (reference environment.py:41-145):
turn-based and simultaneous games, observer players, string-encoded
actions for network matches, and ``diff_info``/``update`` replica sync
for partial-information games.  Environments resolve by short name
through ``ENVS`` or by any dotted module path.

Additions over the reference: ``check_env`` (a lightweight runtime
validator of the contract, useful when wiring a new game) and explicit
contract documentation per method group.
"""

import importlib


ENVS = {
    'TicTacToe':         'handyrl_amd.envs.tictactoe',
    'ParallelTicTacToe': 'handyrl_amd.envs.parallel_tictactoe',
    'Geister':           'handyrl_amd.envs.geister',
    'HungryGeese':       'handyrl_amd.envs.hungry_geese',
}


def _resolve_module(env_args):
    name = env_args['env']
    return importlib.import_module(ENVS.get(name, name))


def prepare_env(env_args):
    """Run an environment module's one-time ``prepare()`` hook if present
    (e.g. asset download / rule-table generation)."""
    module = _resolve_module(env_args)
    hook = getattr(module, 'prepare', None)
    if hook is not None:
        hook()


def make_env(env_args):
    """Instantiate the environment named by ``env_args['env']``."""
    return _resolve_module(env_args).Environment(env_args)


class BaseEnvironment:
    """Abstract game environment.

    A game must implement ``reset``, ``terminal``, ``outcome``,
    ``legal_actions``, ``observation`` and either ``play`` (sequential)
    or ``step`` (simultaneous).  Everything else has a sensible default
    for single-player sequential games.
    """

    def __init__(self, args=None):
        pass

    def __str__(self):
        return ''

    # ---- state transition ----
    # reset() begins a new game; play() applies one player's action;
    # step() applies a dict of simultaneous actions (defaults to playing
    # each non-None entry sequentially).  Transition methods return a
    # truthy value ON FAILURE (the generation loop aborts the episode).

    def reset(self, args=None):
        raise NotImplementedError()

    def play(self, action, player):
        raise NotImplementedError()

    def step(self, actions):
        for player, action in actions.items():
            if action is not None:
                self.play(action, player)

    # ---- whose move ----
    # turn() names the single mover; turns() lists every player acting
    # this step (simultaneous games override it); observers() lists
    # non-acting players that still receive observations (recurrent
    # models keep their hidden state warm through observe steps).

    def turn(self):
        return 0

    def turns(self):
        return [self.turn()]

    def observers(self):
        return []

    # ---- game status ----
    # reward() is the immediate per-player reward of the last transition;
    # outcome() the terminal result per player (zero-sum convention,
    # values in [-1, 1]).

    def terminal(self):
        raise NotImplementedError()

    def reward(self):
        return {}

    def outcome(self):
        raise NotImplementedError()

    def legal_actions(self, player):
        raise NotImplementedError()

    def players(self):
        return [0]

    # ---- encodings ----
    # observation(player) is that player's (partial) view as numpy
    # leaves; action2str/str2action round-trip actions for the network
    # battle protocol.

    def observation(self, player=None):
        raise NotImplementedError()

    def action2str(self, a, player=None):
        return str(a)

    def str2action(self, s, player=None):
        return int(s)

    # ---- replica sync (network matches, partial information) ----
    # diff_info(player) serializes the last transition as seen by that
    # player; update(info, reset) applies it to a replica environment.

    def diff_info(self, player=None):
        return ''

    def update(self, info, reset):
        raise NotImplementedError()


def check_env(env, playouts=3, max_steps=10000):
    """Lightweight runtime validation of the environment contract: runs a
    few random playouts through the full interface and raises AssertionError
    with a readable message on the first violation.  Useful when wiring a
    new game; the env-contract tests use full oracles instead."""
    import random

    players = env.players()
    assert isinstance(players, list) and players, 'players() must be nonempty'
    for _ in range(playouts):
        assert not env.reset(), 'reset() reported failure'
        steps = 0
        while not env.terminal():
            movers = env.turns()
            assert movers, 'turns() empty on a non-terminal state'
            actions = {}
            for p in movers:
                assert p in players, 'turns() outside players()'
                legal = env.legal_actions(p)
                assert len(legal) > 0, 'no legal action for mover %r' % (p,)
                a = random.choice(legal)
                s = env.action2str(a, p)
                assert env.str2action(s, p) == a, \
                    'action2str/str2action round trip failed for %r' % (a,)
                env.observation(p)
                actions[p] = a
            assert not env.step(actions), 'step() reported failure'
            reward = env.reward()
            assert all(p in players for p in reward), 'reward() key outside players()'
            steps += 1
            assert steps < max_steps, 'game did not terminate'
        outcome = env.outcome()
        assert set(outcome) == set(players), 'outcome() must cover players()'
    return True
