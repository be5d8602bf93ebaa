"""Sequential self-play episode generation (CPU worker path).

Parity with reference generation.py: per-player hidden state, per-turn
inference, legal-action masking (+1e32), softmax sampling, per-step moment
records, discounted-return backfill and block-chunked episode packaging.

This is the compatibility path for CPU worker processes; the
high-throughput MI355X path batches many environments through one GPU
forward instead (handyrl_amd/actor.py).
"""

import random

import numpy as np

from .batch import pack_moments
from .util import softmax

MOMENT_KEYS = ('observation', 'selected_prob', 'action_mask', 'action',
               'value', 'reward', 'return')


class Generator:
    def __init__(self, env, args):
        self.env = env
        self.args = args

    def generate(self, models, args):
        moments = []
        hidden = {p: models[p].init_hidden() for p in self.env.players()}

        if self.env.reset():
            return None

        while not self.env.terminal():
            moment = {key: {p: None for p in self.env.players()} for key in MOMENT_KEYS}

            turn_players = self.env.turns()
            observers = self.env.observers()
            for player in self.env.players():
                if player not in turn_players + observers:
                    continue
                if player not in turn_players and player in args['player'] \
                        and not self.args['observation']:
                    continue

                obs = self.env.observation(player)
                outputs = models[player].inference(obs, hidden[player])
                hidden[player] = outputs.get('hidden', None)

                moment['observation'][player] = obs
                moment['value'][player] = outputs.get('value', None)

                if player in turn_players:
                    logits = outputs['policy']
                    legal = self.env.legal_actions(player)
                    action_mask = np.ones_like(logits) * 1e32
                    action_mask[legal] = 0
                    p = softmax(logits - action_mask)
                    action = random.choices(legal, weights=p[legal])[0]

                    moment['selected_prob'][player] = p[action]
                    moment['action_mask'][player] = action_mask
                    moment['action'][player] = action

            if self.env.step(moment['action']):
                return None

            reward = self.env.reward()
            for player in self.env.players():
                moment['reward'][player] = reward.get(player, None)

            moment['turn'] = turn_players
            moments.append(moment)

        if len(moments) < 1:
            return None

        # discounted-return backfill per player
        for player in self.env.players():
            ret = 0
            for i, m in reversed(list(enumerate(moments))):
                ret = (m['reward'][player] or 0) + self.args['gamma'] * ret
                moments[i]['return'][player] = ret

        return {
            'args': args,
            'steps': len(moments),
            'outcome': self.env.outcome(),
            'moment': pack_moments(moments, self.args['compress_steps'],
                                   compress=self.args.get('compress_episodes', True)),
        }

    def execute(self, models, args):
        episode = self.generate(models, args)
        if episode is None:
            print('None episode in generation!')
        return episode
