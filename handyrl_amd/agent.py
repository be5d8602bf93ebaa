"""Agent policies used in matches and evaluation (parity: reference agent.py)."""

import random

import numpy as np

from .util import softmax


class RandomAgent:
    def reset(self, env, show=False):
        pass

    def action(self, env, player, show=False):
        return random.choice(env.legal_actions(player))

    def observe(self, env, player, show=False):
        return [0.0]


class RuleBasedAgent(RandomAgent):
    def __init__(self, key=None):
        self.key = key

    def action(self, env, player, show=False):
        if hasattr(env, 'rule_based_action'):
            return env.rule_based_action(player, key=self.key)
        return random.choice(env.legal_actions(player))


def print_outputs(env, prob, v):
    if hasattr(env, 'print_outputs'):
        env.print_outputs(prob, v)
    else:
        if v is not None:
            print('v = %f' % v)
        if prob is not None:
            print('p = %s' % (prob * 1000).astype(int))


class Agent:
    """Greedy (temperature 0) or temperature-sampling model agent."""

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

    def action(self, env, player, show=False):
        outputs = self.plan(env.observation(player))
        p = outputs['policy']
        v = outputs.get('value', None)
        legal = env.legal_actions(player)
        mask = np.ones_like(p)
        mask[legal] = 0
        p = p - mask * 1e32

        if show:
            print_outputs(env, softmax(p), v)

        if self.temperature == 0:
            return max(legal, key=lambda a: p[a])
        probs = softmax(p / self.temperature)
        return random.choices(np.arange(len(p)), weights=probs)[0]

    def observe(self, env, player, show=False):
        v = None
        if self.observation:
            outputs = self.plan(env.observation(player))
            v = outputs.get('value', None)
            if show:
                print_outputs(env, None, v)
        return v


class EnsembleAgent(Agent):
    """Mean over several models' outputs."""

    def reset(self, env, show=False):
        self.hidden = [model.init_hidden() for model in self.model]

    def plan(self, obs):
        outputs = {}
        for i, model in enumerate(self.model):
            out = model.inference(obs, self.hidden[i])
            for k, v in out.items():
                if k == 'hidden':
                    self.hidden[i] = v
                else:
                    outputs.setdefault(k, []).append(v)
        return {k: np.mean(v, axis=0) for k, v in outputs.items()}


class SoftAgent(Agent):
    def __init__(self, model):
        super().__init__(model, temperature=1.0)
