"""Game-environment contract and registry.

The ``BaseEnvironment`` interface reproduces the reference contract
(reference handyrl/environment.py:41-145): turn-based and simultaneous
games, observer players, string-encoded actions for network matches, and
``diff_info``/``update`` partial-information synchronisation.

Environments are resolved by short name through ``ENVS`` or by an arbitrary
dotted module path (reference environment.py:9-36).
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
    """Run an environment module's one-time ``prepare()`` hook if present."""
    mod = _resolve_module(env_args)
    if hasattr(mod, 'prepare'):
        mod.prepare()


def make_env(env_args):
    """Instantiate the environment described by ``env_args['env']``."""
    mod = _resolve_module(env_args)
    return mod.Environment(env_args)


class BaseEnvironment:
    """Abstract game environment.

    Mandatory for every game: ``reset``, ``terminal``, ``outcome``,
    ``legal_actions``, ``observation`` and either ``play`` (sequential
    games) or ``step`` (simultaneous games).
    """

    def __init__(self, args=None):
        pass

    def __str__(self):
        return ''

    # -- state transition -------------------------------------------------
    def reset(self, args=None):
        raise NotImplementedError()

    def play(self, action, player):
        """Apply a single player's action (sequential games)."""
        raise NotImplementedError()

    def step(self, actions):
        """Apply a dict of simultaneous actions; defaults to sequential play."""
        for p, action in actions.items():
            if action is not None:
                self.play(action, p)

    # -- whose move -------------------------------------------------------
    def turn(self):
        return 0

    def turns(self):
        """Players who act this step (simultaneous games override this)."""
        return [self.turn()]

    def observers(self):
        """Non-acting players that should still observe (e.g. for RNN state)."""
        return []

    # -- game status ------------------------------------------------------
    def terminal(self):
        raise NotImplementedError()

    def reward(self):
        """Immediate per-player rewards for the last transition."""
        return {}

    def outcome(self):
        """Terminal outcome per player (zero-sum convention: in [-1, 1])."""
        raise NotImplementedError()

    def legal_actions(self, player):
        raise NotImplementedError()

    def players(self):
        return [0]

    # -- encodings --------------------------------------------------------
    def observation(self, player=None):
        raise NotImplementedError()

    def action2str(self, a, player=None):
        return str(a)

    def str2action(self, s, player=None):
        return int(s)

    # -- network battle sync ----------------------------------------------
    def diff_info(self, player=None):
        """Serializable description of the last transition, per viewer."""
        return ''

    def update(self, info, reset):
        """Apply ``diff_info`` output to a replica environment."""
        raise NotImplementedError()
