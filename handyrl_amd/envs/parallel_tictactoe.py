"""Simultaneous-move Tic-Tac-Toe (both players submit; one random action lands).

Exercises the simultaneous-transition ``turns()``/``step()`` path of the
environment contract (parity with reference envs/parallel_tictactoe.py).
"""

import random

from .tictactoe import Environment as TicTacToe


class Environment(TicTacToe):

    def __str__(self):
        lines = ['  ' + ' '.join(self.ROWS)]
        for x in range(3):
            row = [self.MARK[int(self.cells[x * 3 + y])] for y in range(3)]
            lines.append(self.COLS[x] + ' ' + ' '.join(row))
        return '\n'.join(lines)

    def step(self, actions):
        chosen = random.choice(list(actions.keys()))
        self._apply(actions[chosen], chosen)

    def _apply(self, action, player):
        mark = 1 if player == 0 else -1
        self.cells[action] = mark
        from .tictactoe import _LINES
        for line in _LINES:
            if action in line and all(self.cells[i] == mark for i in line):
                self.winner = mark
                break
        self.history.append((player, action))

    def diff_info(self, player=None):
        if not self.history:
            return ''
        p, a = self.history[-1]
        return self.action2str(a) + ':' + str(p)

    def update(self, info, reset):
        if reset:
            self.reset()
        else:
            s, p = info.split(':')
            self._apply(self.str2action(s), int(p))

    def turn(self):
        # no single turn player in simultaneous mode; observation() then
        # renders the fixed (non-turn-view) perspective for any player
        return None

    def turns(self):
        return self.players()
