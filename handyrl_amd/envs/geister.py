"""Geister — 6x6 partial-information piece game.

Behavioral parity with reference envs/geister.py: two players each secretly
assign 4 blue / 4 red to 8 fixed start squares (70 layouts), then alternate
single-square moves; win by walking a blue piece off the board through the
opponent-side corner goals, by capturing all opponent blues, or by having
all of your reds captured... from the opponent's perspective: capturing all
of an opponent's RED pieces makes the CAPTURER lose.  200 turns = draw.
Action space: 144 moves (direction*36 + from-square, mover-relative
coordinates) + 70 layout choices.  Per-step reward -0.01; observation is
{scalar(18), board(7,6,6)} rotated to the viewer.
"""

import itertools
import random

import numpy as np

from ..environment import BaseEnvironment
from ..models.geister_net import GeisterNet

BLACK, WHITE = 0, 1
BLUE, RED = 0, 1
BOARD_N = 6
N_MOVE_ACTIONS = 4 * 36          # = 144
N_LAYOUTS = 70

COLS = 'ABCDEF'
ROWS = '123456'
TYPE_CHARS = 'BR'
PIECE_CHARS = {-1: '_', 0: 'B', 1: 'R', 2: 'b', 3: 'r', 4: '*'}

# directions in board coordinates; mover-relative actions rotate for WHITE
DIRS = ((-1, 0), (0, -1), (0, 1), (1, 0))

# start squares, per color, in layout-index order
START_SQUARES = [
    ['B2', 'C2', 'D2', 'E2', 'B1', 'C1', 'D1', 'E1'],
    ['E5', 'D5', 'C5', 'B5', 'E6', 'D6', 'C6', 'B6'],
]

# off-board goal squares per color: (x, y) just beyond the far edge corners
GOALS = (((-1, 5), (6, 5)), ((-1, 0), (6, 0)))

# layout index -> which 4 of the 8 start squares hold BLUE pieces
LAYOUTS = list(itertools.combinations(range(8), 4))


def piece_code(color, ptype):
    return color * 2 + ptype


def piece_color(code):
    return -1 if code < 0 else code // 2


def piece_type(code):
    return -1 if code < 0 else code % 2


def on_board(x, y):
    return 0 <= x < BOARD_N and 0 <= y < BOARD_N


class Environment(BaseEnvironment):

    def __init__(self, args=None):
        super().__init__()
        self.args = args or {}
        self.reset()

    def reset(self, args=None):
        self.board = np.full((BOARD_N, BOARD_N), -1, dtype=np.int32)
        self.color = BLACK
        self.turn_count = -2            # two layout turns precede moves
        self.win_color = None           # None / BLACK / WHITE / 2 (draw)
        self.piece_cnt = np.zeros(4, dtype=np.int32)
        # per piece slot (color*8+idx): (x, y) or (-1, -1) when off board
        self.piece_pos = np.full((16, 2), -1, dtype=np.int32)
        self.slot_of = np.full((BOARD_N, BOARD_N), -1, dtype=np.int32)
        self.record = []
        self.captured_type = None
        self.layouts = {}

    # -- geometry helpers -------------------------------------------------
    @staticmethod
    def _rot(x, y):
        return BOARD_N - 1 - x, BOARD_N - 1 - y

    def _compose_action(self, x, y, d, color):
        if color == WHITE:
            x, y = self._rot(x, y)
            d = 3 - d
        return d * 36 + x * 6 + y

    def _action_from(self, a, color):
        x, y = divmod(a % 36, 6)
        if color == WHITE:
            x, y = self._rot(x, y)
        return x, y

    def _action_dir(self, a, color):
        d = a // 36
        return 3 - d if color == WHITE else d

    def _action_to(self, a, color):
        x, y = self._action_from(a, color)
        dx, dy = DIRS[self._action_dir(a, color)]
        return x + dx, y + dy

    def _sq_str(self, x, y):
        return COLS[x] + ROWS[y] if on_board(x, y) else '**'

    def _sq_parse(self, s):
        if s == '**':
            return None
        return COLS.index(s[0]), ROWS.index(s[1])

    def _is_goal(self, color, x, y):
        return any(gx == x and gy == y for gx, gy in GOALS[color])

    # -- encodings --------------------------------------------------------
    def action2str(self, a, player):
        if a >= N_MOVE_ACTIONS:
            return 's%d' % (a - N_MOVE_ACTIONS)
        fx, fy = self._action_from(a, player)
        tx, ty = self._action_to(a, player)
        return self._sq_str(fx, fy) + self._sq_str(tx, ty)

    def str2action(self, s, player):
        if s[0] == 's':
            return N_MOVE_ACTIONS + int(s[1:])
        fx, fy = self._sq_parse(s[:2])
        to = self._sq_parse(s[2:])
        if to is None:
            # off-board: the adjacent goal square determines the direction
            for gx, gy in GOALS[player]:
                if abs(gx - fx) + abs(gy - fy) == 1:
                    to = (gx, gy)
                    break
        dx, dy = to[0] - fx, to[1] - fy
        d = DIRS.index((dx, dy))
        return self._compose_action(fx, fy, d, player)

    def __str__(self):
        def shown(code):
            if code == -1 or self.layouts.get(piece_color(code), -1) >= 0:
                return code
            return 4
        lines = ['  ' + ' '.join(ROWS)]
        for x in range(BOARD_N):
            row = [PIECE_CHARS[shown(int(self.board[x, y]))] for y in range(BOARD_N)]
            lines.append(COLS[x] + ' ' + ' '.join(row))
        lines.append('remained = B:%d R:%d b:%d r:%d' % tuple(self.piece_cnt))
        lines.append('turn = %-3d color = %s' % (self.turn_count, 'BW'[self.color]))
        return '\n'.join(lines)

    # -- board bookkeeping ------------------------------------------------
    def _place(self, code, x, y, slot):
        self.board[x, y] = code
        self.slot_of[x, y] = slot
        self.piece_pos[slot] = (x, y)
        self.piece_cnt[code] += 1

    def _remove(self, x, y):
        code = int(self.board[x, y])
        slot = int(self.slot_of[x, y])
        self.board[x, y] = -1
        self.slot_of[x, y] = -1
        self.piece_pos[slot] = (-1, -1)
        self.piece_cnt[code] -= 1
        return code

    def _relocate(self, fx, fy, tx, ty):
        code = int(self.board[fx, fy])
        slot = int(self.slot_of[fx, fy])
        self.board[fx, fy] = -1
        self.slot_of[fx, fy] = -1
        self.board[tx, ty] = code
        self.slot_of[tx, ty] = slot
        self.piece_pos[slot] = (tx, ty)

    def _apply_layout(self, layout):
        self.layouts[self.color] = layout
        if layout < 0:
            layout = random.randrange(N_LAYOUTS)
        blues = LAYOUTS[layout]
        for idx, sq in enumerate(START_SQUARES[self.color]):
            ptype = BLUE if idx in blues else RED
            x, y = self._sq_parse(sq)
            self._place(piece_code(self.color, ptype), x, y, self.color * 8 + idx)
        self.color ^= 1
        self.turn_count += 1

    # -- transitions ------------------------------------------------------
    def play(self, action, player=None):
        if self.turn_count < 0:
            return self._apply_layout(action - N_MOVE_ACTIONS)

        me = self.color
        fx, fy = self._action_from(action, me)
        tx, ty = self._action_to(action, me)
        self.captured_type = None

        if not on_board(tx, ty):
            # a blue walks through the goal: immediate win
            self._remove(fx, fy)
            self.win_color = me
        else:
            target = int(self.board[tx, ty])
            if target != -1:
                self._remove(tx, ty)
                self.captured_type = piece_type(target)
                if self.piece_cnt[target] == 0:
                    if piece_type(target) == BLUE:
                        self.win_color = me            # all enemy blues captured
                    else:
                        self.win_color = me ^ 1        # captured all enemy reds: lose
            self._relocate(fx, fy, tx, ty)

        self.color ^= 1
        self.turn_count += 1
        self.record.append(action)

        if self.turn_count >= 200 and self.win_color is None:
            self.win_color = 2

    # -- network-battle sync ----------------------------------------------
    def diff_info(self, player):
        mover = (self.turn_count - 1) % 2
        info = {}
        if len(self.record) == 0:
            if self.turn_count > -2:
                info['set'] = self.layouts[mover] if player == mover else -1
        else:
            info['move'] = self.action2str(self.record[-1], mover)
            if player == mover and self.captured_type is not None:
                info['captured'] = TYPE_CHARS[self.captured_type]
        return info

    def update(self, info, reset):
        if reset:
            self.reset(info)
        elif 'set' in info:
            self._apply_layout(info['set'])
        elif 'move' in info:
            action = self.str2action(info['move'], self.color)
            if 'captured' in info:
                # reveal the true type of the piece about to be captured:
                # repaint the board cell only (piece counts stay keyed to the
                # true types, which every reveal decrements consistently)
                tx, ty = self._action_to(action, self.color)
                t = TYPE_CHARS.index(info['captured'])
                self.board[tx, ty] = piece_code(self.color ^ 1, t)
            self.play(action)

    # -- status -----------------------------------------------------------
    def turn(self):
        return self.players()[self.turn_count % 2]

    def terminal(self):
        return self.win_color is not None

    def reward(self):
        return {p: -0.01 for p in self.players()}

    def outcome(self):
        if self.win_color == BLACK:
            oc = [1, -1]
        elif self.win_color == WHITE:
            oc = [-1, 1]
        else:
            oc = [0, 0]
        return {p: oc[i] for i, p in enumerate(self.players())}

    # -- legality ---------------------------------------------------------
    def _move_ok(self, color, ptype, tx, ty):
        if on_board(tx, ty):
            return piece_color(int(self.board[tx, ty])) != color
        return ptype == BLUE and self._is_goal(color, tx, ty)

    def legal(self, action):
        if self.turn_count < 0:
            return 0 <= action - N_MOVE_ACTIONS < N_LAYOUTS
        if not 0 <= action < N_MOVE_ACTIONS:
            return False
        fx, fy = self._action_from(action, self.color)
        code = int(self.board[fx, fy])
        if piece_color(code) != self.color:
            return False
        tx, ty = self._action_to(action, self.color)
        return self._move_ok(self.color, piece_type(code), tx, ty)

    def legal_actions(self, player=None):
        if self.turn_count < 0:
            return [N_MOVE_ACTIONS + i for i in range(N_LAYOUTS)]
        actions = []
        me = self.color
        for slot in range(me * 8, me * 8 + 8):
            x, y = self.piece_pos[slot]
            if x < 0:
                continue
            ptype = piece_type(int(self.board[x, y]))
            for d, (dx, dy) in enumerate(DIRS):
                if self._move_ok(me, ptype, x + dx, y + dy):
                    actions.append(self._compose_action(x, y, d, me))
        return actions

    def players(self):
        return [0, 1]

    # -- learning interface -----------------------------------------------
    def net(self):
        return GeisterNet()

    def observation(self, player=None):
        turn_view = player is None or player == self.turn()
        me = self.color if turn_view else self.color ^ 1
        opp = me ^ 1

        def count_onehot(code):
            n = int(self.piece_cnt[code])
            return [1.0 if n == i else 0.0 for i in range(1, 5)]

        scalar = np.array(
            [1.0 if me == BLACK else 0.0, 1.0 if turn_view else 0.0]
            + count_onehot(piece_code(me, BLUE)) + count_onehot(piece_code(me, RED))
            + count_onehot(piece_code(opp, BLUE)) + count_onehot(piece_code(opp, RED)),
            dtype=np.float32)

        my_blue = self.board == piece_code(me, BLUE)
        my_red = self.board == piece_code(me, RED)
        opp_blue = self.board == piece_code(opp, BLUE)
        opp_red = self.board == piece_code(opp, RED)
        full_view = player is None
        zeros = np.zeros_like(self.board, dtype=bool)

        planes = np.stack([
            np.ones_like(self.board),
            my_blue + my_red,
            opp_blue + opp_red,
            my_blue,
            my_red,
            opp_blue if full_view else zeros,
            opp_red if full_view else zeros,
        ]).astype(np.float32)

        if me == WHITE:
            planes = np.rot90(planes, k=2, axes=(1, 2)).copy()

        return {'scalar': scalar, 'board': planes}


if __name__ == '__main__':
    e = Environment()
    for _ in range(5):
        e.reset()
        while not e.terminal():
            e.play(random.choice(e.legal_actions()))
        print(e)
        print(e.outcome())
