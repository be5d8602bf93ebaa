"""Hungry Geese — self-contained reimplementation (no kaggle_environments).

Rules follow the Kaggle "hungry_geese" environment that the reference wraps
(reference envs/kaggle/hungry_geese.py): a 7x11 torus, 4 geese, 2 food on
the board, reverse-move and collision deaths, tail-shrink every
``HUNGER_RATE`` steps, 200-step limit, rank-based outcome.  The 17-plane
observation encoding and GeeseNet (torus-conv residual tower) match the
reference wrapper (envs/kaggle/hungry_geese.py:23-57, 202-231).

The per-goose running score is ``steps_survived * (MAX_LEN + 1) + length``
so final ranking is (survival time, then length), as in the Kaggle env.
"""

import random

import numpy as np
import torch
import torch.nn as nn
import torch.nn.functional as F

from ..environment import BaseEnvironment
from ..models.geese_net import GeeseNet

ROWS, COLS = 7, 11
N_CELLS = ROWS * COLS
N_PLAYERS = 4
HUNGER_RATE = 40
MIN_FOOD = 2
MAX_LEN = 99
MAX_STEPS = 199          # transitions; episode length cap

# action index -> (drow, dcol); OPPOSITE[a] is the reverse action
ACTION_NAMES = ['NORTH', 'SOUTH', 'WEST', 'EAST']
MOVES = [(-1, 0), (1, 0), (0, -1), (0, 1)]
OPPOSITE = [1, 0, 3, 2]


def shift(pos, action):
    r, c = divmod(pos, COLS)
    dr, dc = MOVES[action]
    return ((r + dr) % ROWS) * COLS + (c + dc) % COLS


class GeeseState:
    """Plain game state; stepped in-place. Separated from the Environment
    wrapper so the vectorized self-play engine can share the rule logic."""

    def __init__(self, rng=None):
        self.rng = rng or random
        self.reset()

    def reset(self):
        cells = self.rng.sample(range(N_CELLS), N_PLAYERS + MIN_FOOD)
        self.geese = [[c] for c in cells[:N_PLAYERS]]
        self.food = set(cells[N_PLAYERS:])
        self.alive = [True] * N_PLAYERS
        self.scores = [0.0] * N_PLAYERS
        self.last_actions = [None] * N_PLAYERS
        self.prev_heads = [None] * N_PLAYERS
        self.step_count = 0
        self.over = False

    def _kill(self, p):
        self.alive[p] = False
        self.geese[p] = []

    def step(self, actions):
        """Apply one simultaneous transition. ``actions``: {player: 0..3}."""
        self.prev_heads = [g[0] if g else None for g in self.geese]

        # move phase (player-index order, as the Kaggle interpreter does)
        for p in range(N_PLAYERS):
            if not self.alive[p]:
                continue
            a = actions.get(p)
            if a is None:
                a = 0
            last = self.last_actions[p]
            if last is not None and a == OPPOSITE[last]:
                self._kill(p)
                continue
            self.last_actions[p] = a
            goose = self.geese[p]
            head = shift(goose[0], a)
            if head in self.food:
                self.food.discard(head)
            else:
                goose.pop()
            goose.insert(0, head)
            # hunger: shrink every HUNGER_RATE transitions
            if (self.step_count + 1) % HUNGER_RATE == 0:
                if goose:
                    goose.pop()
                if not goose:
                    self._kill(p)

        # collision phase: any head sharing a cell with anything dies
        occupancy = {}
        for g in self.geese:
            for cell in g:
                occupancy[cell] = occupancy.get(cell, 0) + 1
        for p in range(N_PLAYERS):
            if self.alive[p] and occupancy.get(self.geese[p][0], 0) > 1:
                self._kill(p)

        # food replenishment
        need = MIN_FOOD - len(self.food)
        if need > 0:
            taken = set(self.food)
            for g in self.geese:
                taken.update(g)
            free = [c for c in range(N_CELLS) if c not in taken]
            if free:
                for c in self.rng.sample(free, min(need, len(free))):
                    self.food.add(c)

        self.step_count += 1

        # scoring for survivors
        for p in range(N_PLAYERS):
            if self.alive[p]:
                self.scores[p] = self.step_count * (MAX_LEN + 1) + len(self.geese[p])

        # termination: <=1 goose left, or step limit
        if sum(self.alive) <= 1 or self.step_count >= MAX_STEPS:
            self.over = True


class Environment(BaseEnvironment):
    ACTION = ACTION_NAMES

    def __init__(self, args=None):
        super().__init__()
        args = args or {}
        seed = args.get('id')
        self.rng = random.Random(seed) if seed is not None else random
        self.state = GeeseState(self.rng)

    def reset(self, args=None):
        self.state.reset()

    # -- encodings --------------------------------------------------------
    def action2str(self, a, player=None):
        return self.ACTION[a]

    def str2action(self, s, player=None):
        return self.ACTION.index(s)

    def __str__(self):
        st = self.state
        grid = ['.'] * N_CELLS
        for c in st.food:
            grid[c] = 'f'
        for p, g in enumerate(st.geese):
            for c in g:
                grid[c] = str(p)
            if g:
                grid[g[0]] = 'ABCD'[p]
        rows = [''.join(grid[r * COLS:(r + 1) * COLS]) for r in range(ROWS)]
        status = ' '.join('%d:%s' % (p, len(g) if st.alive[p] else '-')
                          for p, g in enumerate(st.geese))
        return 'step %d\n%s\n%s' % (st.step_count, '\n'.join(rows), status)

    # -- transitions ------------------------------------------------------
    def step(self, actions):
        self.state.step({p: a for p, a in actions.items() if a is not None})

    def diff_info(self, player=None):
        """Full-information game: ship the whole state snapshot."""
        st = self.state
        return {
            'geese': [list(g) for g in st.geese],
            'food': sorted(st.food),
            'alive': list(st.alive),
            'scores': list(st.scores),
            'last_actions': list(st.last_actions),
            'prev_heads': list(st.prev_heads),
            'step_count': st.step_count,
            'over': st.over,
        }

    def update(self, info, reset):
        st = self.state
        st.geese = [list(g) for g in info['geese']]
        st.food = set(info['food'])
        st.alive = list(info['alive'])
        st.scores = list(info['scores'])
        st.last_actions = list(info['last_actions'])
        st.prev_heads = list(info['prev_heads'])
        st.step_count = info['step_count']
        st.over = info['over']

    # -- status -----------------------------------------------------------
    def turns(self):
        if self.state.over:
            return []
        return [p for p in self.players() if self.state.alive[p]]

    def terminal(self):
        return self.state.over

    def outcome(self):
        """Pairwise rank scoring in [-1, 1] (reference wrapper :168-180)."""
        sc = self.state.scores
        outcomes = {}
        for p in self.players():
            o = 0.0
            for q in self.players():
                if p == q:
                    continue
                if sc[p] > sc[q]:
                    o += 1 / (N_PLAYERS - 1)
                elif sc[p] < sc[q]:
                    o -= 1 / (N_PLAYERS - 1)
            outcomes[p] = o
        return outcomes

    def legal_actions(self, player):
        return list(range(4))

    def players(self):
        return list(range(N_PLAYERS))

    def rule_based_action(self, player, key=None):
        """Greedy baseline: head toward nearest food, avoiding immediate
        death (occupied cells and the reverse move)."""
        st = self.state
        if not st.alive[player]:
            return 0
        head = st.geese[player][0]
        occupied = set()
        for g in st.geese:
            occupied.update(g[:-1] if len(g) > 1 else g)  # tails will move
        banned = OPPOSITE[st.last_actions[player]] if st.last_actions[player] is not None else None

        def dist(c):
            if not st.food:
                return 0
            r, col = divmod(c, COLS)
            best = N_CELLS
            for f in st.food:
                fr, fc = divmod(f, COLS)
                dr = min(abs(fr - r), ROWS - abs(fr - r))
                dc = min(abs(fc - col), COLS - abs(fc - col))
                best = min(best, dr + dc)
            return best

        candidates = []
        for a in range(4):
            if a == banned:
                continue
            nxt = shift(head, a)
            candidates.append((nxt in occupied, dist(nxt), a))
        if not candidates:
            return 0
        candidates.sort()
        return candidates[0][2]

    # -- learning interface -----------------------------------------------
    def net(self):
        return GeeseNet()

    def observation(self, player=None):
        """17 planes of 7x11, player-relative channel assignment:
        [0:4] heads, [4:8] tails, [8:12] bodies, [12:16] previous heads,
        [16] food (reference wrapper :202-231)."""
        if player is None:
            player = 0
        st = self.state
        planes = np.zeros((N_PLAYERS * 4 + 1, N_CELLS), dtype=np.float32)
        for p, g in enumerate(st.geese):
            rel = (p - player) % N_PLAYERS
            if g:
                planes[rel, g[0]] = 1
                planes[4 + rel, g[-1]] = 1
                for c in g:
                    planes[8 + rel, c] = 1
        for p, h in enumerate(st.prev_heads):
            if h is not None:
                planes[12 + (p - player) % N_PLAYERS, h] = 1
        for c in st.food:
            planes[16, c] = 1
        return planes.reshape(-1, ROWS, COLS)


if __name__ == '__main__':
    e = Environment()
    for _ in range(3):
        e.reset()
        while not e.terminal():
            e.step({p: e.rule_based_action(p) for p in e.turns()})
        print(e)
        print(e.outcome())
