"""Sequential self-play episode generation (CPU worker path).

Produces reference-format episodes (reference generation.py:20-93): one
moment dict per step with per-player observation / selected_prob /
action_mask / action / value / reward / return entries, packaged in
``compress_steps`` blocks.  The loop itself is organized differently from
the reference — per-step records are built by small helper stages and the
discounted returns are computed from a reward matrix in one vectorized
backward scan — but the emitted episode is interchangeable.

This is the compatibility path for CPU worker processes; the
high-throughput MI355X path batches thousands of environments through one
GPU forward instead (handyrl_amd/actor.py).
"""

import random

import numpy as np

from .batch import pack_moments
from .util import softmax

MOMENT_KEYS = ('observation', 'selected_prob', 'action_mask', 'action',
               'value', 'reward', 'return')


class Generator:
    """Runs one self-play episode and packages it for the replay buffer."""

    def __init__(self, env, args):
        self.env = env
        self.args = args

    # -- per-step helpers ---------------------------------------------------

    def _active_players(self, turn_players, observers, my_players):
        """Players whose model runs this step: everyone acting, plus
        observers, minus our own non-acting seats when observation
        inference is disabled."""
        active = []
        for p in self.env.players():
            if p in turn_players:
                active.append(p)
            elif p in observers:
                if p in my_players and not self.args['observation']:
                    continue
                active.append(p)
        return active

    def _sample_action(self, logits, legal):
        """Masked softmax draw; returns (action, behavior_prob, mask)."""
        mask = np.full_like(logits, 1e32)
        mask[legal] = 0
        probs = softmax(logits - mask)
        action = random.choices(legal, weights=probs[legal])[0]
        return action, probs[action], mask

    # -- episode loop -------------------------------------------------------

    def generate(self, models, args):
        env, players = self.env, self.env.players()
        hidden = {p: models[p].init_hidden() for p in players}
        if env.reset():
            return None

        records = []                       # one dict per step
        reward_rows = []                   # (step, player) reward matrix

        while not env.terminal():
            turn_players = env.turns()
            rec = {key: dict.fromkeys(players) for key in MOMENT_KEYS}
            rec['turn'] = turn_players

            for p in self._active_players(turn_players, env.observers(),
                                          args['player']):
                obs = env.observation(p)
                out = models[p].inference(obs, hidden[p])
                hidden[p] = out.get('hidden')
                rec['observation'][p] = obs
                rec['value'][p] = out.get('value')
                if p in turn_players:
                    a, prob, mask = self._sample_action(out['policy'],
                                                        env.legal_actions(p))
                    rec['action'][p] = a
                    rec['selected_prob'][p] = prob
                    rec['action_mask'][p] = mask

            if env.step(rec['action']):
                return None
            step_reward = env.reward()
            for p in players:
                rec['reward'][p] = step_reward.get(p)
            reward_rows.append([step_reward.get(p) or 0 for p in players])
            records.append(rec)

        if not records:
            return None

        # discounted-return backfill, one vectorized scan over the matrix
        rew = np.asarray(reward_rows, dtype=np.float64)     # (T, P)
        ret = np.zeros(len(players), dtype=np.float64)
        for t in range(len(records) - 1, -1, -1):
            ret = rew[t] + self.args['gamma'] * ret
            for i, p in enumerate(players):
                records[t]['return'][p] = ret[i]

        return {
            'args': args,
            'steps': len(records),
            'outcome': env.outcome(),
            'moment': pack_moments(
                records, self.args['compress_steps'],
                compress=self.args.get('compress_episodes', True)),
        }

    def execute(self, models, args):
        episode = self.generate(models, args)
        if episode is None:
            print('None episode in generation!')
        return episode
