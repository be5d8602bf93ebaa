"""Agent policies used in matches and evaluation.

Semantics contract (reference agent.py): ``Agent`` plays greedily at
temperature 0 and samples a Boltzmann policy otherwise; ``SoftAgent`` is
temperature 1; ``RandomAgent`` uniform over legal actions;
``RuleBasedAgent`` defers to ``env.rule_based_action``; ``EnsembleAgent``
averages several models' outputs.  Observing players run value-only
inference to keep recurrent hidden state warm.
"""

import random

import numpy as np

from .util import softmax


def _legal_logits(policy, legal):
    """Push illegal actions to -1e32 (the framework-wide masking scheme)."""
    masked = policy - 1e32
    masked[legal] = policy[legal]
    return masked


def print_outputs(env, prob, v):
    if hasattr(env, 'print_outputs'):
        env.print_outputs(prob, v)
        return
    if v is not None:
        print('v = %f' % v)
    if prob is not None:
        print('p = %s' % (prob * 1000).astype(int))


class RandomAgent:
    """Uniform over legal actions; no model, no state."""

    def reset(self, env, show=False):
        pass

    def action(self, env, player, show=False):
        return random.choice(env.legal_actions(player))

    def observe(self, env, player, show=False):
        return [0.0]


class RuleBasedAgent(RandomAgent):
    """Delegates to the environment's built-in heuristic when it has one."""

    def __init__(self, key=None):
        self.key = key

    def action(self, env, player, show=False):
        rule = getattr(env, 'rule_based_action', None)
        if rule is None:
            return random.choice(env.legal_actions(player))
        return rule(player, key=self.key)


class Agent:
    """Model-driven agent.

    ``temperature == 0`` -> argmax over legal logits; otherwise a sample
    from softmax(logits / temperature).  ``observation=True`` keeps
    running inference on observe() steps so recurrent hidden state tracks
    the game even when it is not this agent's turn.
    """

    def __init__(self, model, temperature=0.0, observation=True):
        self.model = model
        self.hidden = None
        self.temperature = temperature
        self.observation = observation

    def reset(self, env, show=False):
        self.hidden = self.model.init_hidden()

    def plan(self, obs):
        outputs = self.model.inference(obs, self.hidden)
        self.hidden = outputs.pop('hidden', None)
        return outputs

    def _choose(self, logits, legal):
        if self.temperature == 0:
            return max(legal, key=lambda a: logits[a])
        weights = softmax(logits / self.temperature)
        return random.choices(range(len(logits)), weights=weights)[0]

    def action(self, env, player, show=False):
        outputs = self.plan(env.observation(player))
        legal = env.legal_actions(player)
        logits = _legal_logits(np.asarray(outputs['policy'], dtype=np.float64),
                               legal)
        if show:
            print_outputs(env, softmax(logits), outputs.get('value'))
        return self._choose(logits, legal)

    def observe(self, env, player, show=False):
        if not self.observation:
            return None
        outputs = self.plan(env.observation(player))
        v = outputs.get('value')
        if show:
            print_outputs(env, None, v)
        return v


class EnsembleAgent(Agent):
    """Arithmetic mean of several models' outputs (self.model is a list)."""

    def reset(self, env, show=False):
        self.hidden = [m.init_hidden() for m in self.model]

    def plan(self, obs):
        pooled = {}
        for i, model in enumerate(self.model):
            out = model.inference(obs, self.hidden[i])
            self.hidden[i] = out.pop('hidden', None)
            for k, v in out.items():
                pooled.setdefault(k, []).append(v)
        return {k: np.mean(vs, axis=0) for k, vs in pooled.items()}


class SoftAgent(Agent):
    """Temperature-1 sampling agent (exploration-shaped evaluation)."""

    def __init__(self, model):
        super().__init__(model, temperature=1.0)
