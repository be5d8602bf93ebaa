"""Tic-Tac-Toe environment and its small conv policy-value net.

Behavioral parity with reference handyrl/envs/tictactoe.py (3x3 alternating
game, observation 3x3x3, 'A1'-style action strings, SimpleConv2dModel with
3 BN conv blocks and conv+FC heads).  Implementation is original.
"""

import random

import numpy as np
import torch
import torch.nn as nn
import torch.nn.functional as F

from ..environment import BaseEnvironment
from ..models.common import apply_bn

# the eight winning index-triples of a 3x3 board (cells are x*3+y)
_LINES = [
    (0, 1, 2), (3, 4, 5), (6, 7, 8),     # rows
    (0, 3, 6), (1, 4, 7), (2, 5, 8),     # cols
    (0, 4, 8), (2, 4, 6),                # diagonals
]


class ConvBN(nn.Module):
    """3x3 conv, optional BatchNorm (bias dropped when BN is on)."""

    def __init__(self, ch_in, ch_out, ksize=3, bn=True):
        super().__init__()
        self.conv = nn.Conv2d(ch_in, ch_out, ksize, padding=ksize // 2, bias=not bn)
        self.bn = nn.BatchNorm2d(ch_out) if bn else None

    def forward(self, x):
        return apply_bn(self.bn, self.conv(x))


class ConvHead(nn.Module):
    """1x1 conv -> LeakyReLU -> fully-connected head (reference
    tictactoe.py:35-49).  The 1x1 conv nests as ``conv.conv`` so the
    state_dict layout matches reference checkpoints
    (tests/test_checkpoint_compat.py)."""

    def __init__(self, shape, mid_filters, outputs):
        super().__init__()
        ch, hh, ww = shape
        self.conv = ConvBN(ch, mid_filters, ksize=1, bn=False)
        self.fc = nn.Linear(mid_filters * hh * ww, outputs, bias=False)

    def forward(self, x):
        h = F.leaky_relu(self.conv(x), 0.1)
        return self.fc(h.flatten(1))


class SimpleConv2dModel(nn.Module):
    """Conv tower for 3x3 boards: stem + 3 BN blocks + policy/value heads."""

    def __init__(self, ch_in=3, filters=32, blocks=3, actions=9):
        super().__init__()
        # attribute named 'conv' for reference-checkpoint compatibility
        self.conv = nn.Conv2d(ch_in, filters, 3, padding=1)
        self.blocks = nn.ModuleList(ConvBN(filters, filters) for _ in range(blocks))
        self.head_p = ConvHead((filters, 3, 3), 2, actions)
        self.head_v = ConvHead((filters, 3, 3), 1, 1)

    def forward(self, x, hidden=None):
        h = F.relu(self.conv(x))
        for blk in self.blocks:
            h = F.relu(blk(h))
        return {'policy': self.head_p(h), 'value': torch.tanh(self.head_v(h))}


class Environment(BaseEnvironment):
    COLS = 'ABC'
    ROWS = '123'
    MARK = {0: '_', 1: 'O', -1: 'X'}

    def __init__(self, args=None):
        super().__init__()
        self.reset()

    def reset(self, args=None):
        self.cells = np.zeros(9, dtype=np.int64)  # +1 first player, -1 second
        self.to_move = 1
        self.winner = 0          # +1 / -1 / 0 (none yet or draw)
        self.history = []

    # -- encodings --------------------------------------------------------
    def action2str(self, a, player=None):
        return self.COLS[a // 3] + self.ROWS[a % 3]

    def str2action(self, s, player=None):
        return self.COLS.index(s[0]) * 3 + self.ROWS.index(s[1])

    def __str__(self):
        lines = ['  ' + ' '.join(self.ROWS)]
        for x in range(3):
            row = [self.MARK[int(self.cells[x * 3 + y])] for y in range(3)]
            lines.append(self.COLS[x] + ' ' + ' '.join(row))
        lines.append('record = ' + ' '.join(self.action2str(a) for a in self.history))
        return '\n'.join(lines)

    # -- transitions ------------------------------------------------------
    def play(self, action, player=None):
        mark = self.to_move
        self.cells[action] = mark
        for line in _LINES:
            if action in line and all(self.cells[i] == mark for i in line):
                self.winner = mark
                break
        self.history.append(action)
        self.to_move = -mark

    def diff_info(self, player=None):
        return self.action2str(self.history[-1]) if self.history else ''

    def update(self, info, reset):
        if reset:
            self.reset()
        else:
            self.play(self.str2action(info))

    # -- status -----------------------------------------------------------
    def turn(self):
        return len(self.history) % 2

    def terminal(self):
        return self.winner != 0 or len(self.history) == 9

    def outcome(self):
        if self.winner > 0:
            oc = [1, -1]
        elif self.winner < 0:
            oc = [-1, 1]
        else:
            oc = [0, 0]
        return {p: oc[i] for i, p in enumerate(self.players())}

    def legal_actions(self, player=None):
        return [a for a in range(9) if self.cells[a] == 0]

    def players(self):
        return [0, 1]

    # -- learning interface -----------------------------------------------
    def net(self):
        return SimpleConv2dModel()

    def observation(self, player=None):
        """3 planes: is-it-my-turn-view flag, my marks, opponent marks."""
        turn_view = player is None or player == self.turn()
        mine = self.to_move if turn_view else -self.to_move
        board = self.cells.reshape(3, 3)
        return np.stack([
            np.full((3, 3), 1.0 if turn_view else 0.0),
            (board == mine).astype(np.float32),
            (board == -mine).astype(np.float32),
        ]).astype(np.float32)


if __name__ == '__main__':
    e = Environment()
    for _ in range(20):
        e.reset()
        while not e.terminal():
            e.play(random.choice(e.legal_actions()))
        print(e)
        print(e.outcome())
