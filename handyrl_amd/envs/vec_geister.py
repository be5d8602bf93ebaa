"""Vectorized Geister engine: G games stepped at once in numpy.

Same rules as handyrl_amd.envs.geister.Environment (the single-game
oracle; parity is tested in tests/test_vec_geister.py), restructured as
struct-of-arrays so the Geister actor pool (handyrl_amd/actor_geister.py)
can run hundreds of self-play games against ONE batched DRC forward per
turn — the MI355X replacement for the reference's one-process-per
-environment workers (reference worker.py / generation.py).  Rule
semantics follow reference handyrl/envs/geister.py: play()
geister.py:361-439, legal_actions() :476-490, observation() :495-522,
outcome() :440-447 — via the local single-game oracle
handyrl_amd/envs/geister.py, which those parity tests compare against.

State per game: a flat 36-cell board of piece codes (-1 empty, color*2 +
type), a 16-slot piece position table, per-code piece counts, mover color
and turn counter.  All 144 move actions are resolved through precomputed
(color, action) -> from-cell / to-cell / goal tables, so legality, moves,
captures and win detection are fancy-indexed array ops with no per-game
python.
"""

import itertools
import os

import numpy as np


def _load_native_core():
    """In-tree C++ env core (envs/src/vec_geister_core.cpp): legality,
    observation build and step as native per-game loops, bit-equal to the
    numpy engine (tests/test_vec_geister_native.py).
    HANDYRL_NO_NATIVE_ENV=1 or a missing .so falls back to numpy."""
    if os.environ.get('HANDYRL_NO_NATIVE_ENV') == '1':
        return None
    so = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                      '_vec_geister_core.so')
    if not os.path.exists(so):
        return None
    try:
        import importlib.machinery
        import importlib.util
        loader = importlib.machinery.ExtensionFileLoader(
            '_vec_geister_core', so)
        spec = importlib.util.spec_from_loader('_vec_geister_core', loader)
        mod = importlib.util.module_from_spec(spec)
        loader.exec_module(mod)
        return mod
    except Exception:            # noqa: BLE001 - numpy fallback
        return None

BLACK, WHITE = 0, 1
BLUE, RED = 0, 1
BOARD_N = 6
N_CELLS = 36
N_MOVE_ACTIONS = 144
N_LAYOUTS = 70
N_ACTIONS = N_MOVE_ACTIONS + N_LAYOUTS          # 214
MAX_TURNS = 200
ILLEGAL = np.float32(1e32)

# board coordinates: cell = x * 6 + y (x indexes columns A-F, y rows 1-6)
_DIRS = np.array([(-1, 0), (0, -1), (0, 1), (1, 0)], dtype=np.int64)

# (color, action) tables for the 144 move actions: a = d*36 + x*6 + y in
# MOVER-RELATIVE coordinates; WHITE's view is the 180-degree rotation
_ax, _ay = np.divmod(np.arange(N_MOVE_ACTIONS) % 36, 6)
_ad = np.arange(N_MOVE_ACTIONS) // 36
_FX = np.stack([_ax, 5 - _ax])                  # (2, 144) from-x per color
_FY = np.stack([_ay, 5 - _ay])
_D = np.stack([_ad, 3 - _ad])
_TX = _FX + _DIRS[_D, 0]
_TY = _FY + _DIRS[_D, 1]
_ONB = (_TX >= 0) & (_TX < 6) & (_TY >= 0) & (_TY < 6)
A_FROM = (_FX * 6 + _FY).astype(np.int64)       # (2, 144) from cell
A_TO = np.where(_ONB, _TX * 6 + _TY, -1).astype(np.int64)   # -1 = off board

# off-board goal squares per color (just beyond the far-edge corners)
_GOALS = (((-1, 5), (6, 5)), ((-1, 0), (6, 0)))
A_GOAL = np.zeros((2, N_MOVE_ACTIONS), dtype=bool)
for _c, _gs in enumerate(_GOALS):
    for _gx, _gy in _gs:
        A_GOAL[_c] |= (_TX[_c] == _gx) & (_TY[_c] == _gy)

# layout index -> blue positions among the 8 start squares
LAYOUT_BLUE = np.zeros((N_LAYOUTS, 8), dtype=bool)
for _l, _blues in enumerate(itertools.combinations(range(8), 4)):
    LAYOUT_BLUE[_l, list(_blues)] = True

# start squares per color in layout-index order (geister.py START_SQUARES)
_SQ = {'A': 0, 'B': 1, 'C': 2, 'D': 3, 'E': 4, 'F': 5}
START_CELLS = np.array([
    [_SQ[s[0]] * 6 + int(s[1]) - 1 for s in row] for row in (
        ['B2', 'C2', 'D2', 'E2', 'B1', 'C1', 'D1', 'E1'],
        ['E5', 'D5', 'C5', 'B5', 'E6', 'D6', 'C6', 'B6'])],
    dtype=np.int64)                             # (2, 8)

_CORE = _load_native_core()
if _CORE is not None:
    _CORE.set_tables(np.ascontiguousarray(A_FROM),
                     np.ascontiguousarray(A_TO),
                     np.ascontiguousarray(A_GOAL),
                     np.ascontiguousarray(LAYOUT_BLUE),
                     np.ascontiguousarray(START_CELLS))


class GeisterVecEnv:
    """G simultaneous Geister games (layout turns included: actions
    144..213 choose the mover's secret arrangement)."""

    def __init__(self, n_games, seed=0):
        self.G = n_games
        self.rng = np.random.default_rng(seed)
        G = n_games
        self.board = np.full((G, N_CELLS), -1, dtype=np.int8)
        self.slot_of = np.full((G, N_CELLS), -1, dtype=np.int8)
        self.piece_pos = np.full((G, 16), -1, dtype=np.int8)
        self.piece_cnt = np.zeros((G, 4), dtype=np.int8)
        self.color = np.zeros(G, dtype=np.int64)
        self.turn_count = np.full(G, -2, dtype=np.int16)
        self.win = np.full(G, -1, dtype=np.int8)   # -1 none / 0 / 1 / 2 draw
        self.over = np.zeros(G, dtype=bool)

    def reset_games(self, games):
        """Reset the given game indices to the pre-layout state."""
        if len(games) == 0:
            return
        self.board[games] = -1
        self.slot_of[games] = -1
        self.piece_pos[games] = -1
        self.piece_cnt[games] = 0
        self.color[games] = BLACK
        self.turn_count[games] = -2
        self.win[games] = -1
        self.over[games] = False

    def turn(self):
        """Player to move per game (equals mover color)."""
        return self.color

    # -- legality ---------------------------------------------------------
    def legal_masks(self, out=None):
        """float32 (G, 214) additive masks: 0 at legal actions, 1e32
        elsewhere (finished games get all-illegal rows)."""
        G = self.G
        mask = out if out is not None else \
            np.empty((G, N_ACTIONS), dtype=np.float32)
        if _CORE is not None and mask.flags.c_contiguous:
            _CORE.legal_masks_core(self.board, self.color, self.turn_count,
                                   self.win, mask)
            return mask
        mask[:] = ILLEGAL
        live = self.win < 0
        lay = live & (self.turn_count < 0)
        mask[lay, N_MOVE_ACTIONS:] = 0.0
        gm = np.nonzero(live & ~lay)[0]
        if len(gm):
            c = self.color[gm]                         # (n,)
            fcode = self.board[gm[:, None], A_FROM[c]]  # (n, 144)
            mine = (fcode >= 0) & ((fcode >> 1) == c[:, None])
            to = A_TO[c]
            tcode = self.board[gm[:, None], np.maximum(to, 0)]
            ok_on = (to >= 0) & ~((tcode >= 0) & ((tcode >> 1) == c[:, None]))
            ok_goal = A_GOAL[c] & ((fcode & 1) == BLUE)
            legal = mine & (ok_on | ok_goal)
            mask[gm, :N_MOVE_ACTIONS] = np.where(legal, np.float32(0), ILLEGAL)
        return mask

    # -- observation ------------------------------------------------------
    def observations(self):
        """Current-mover observations (partial view, mover-rotated):
        scalar (G, 18) float32 + board planes (G, 7, 6, 6) float32 —
        geister.py Environment.observation(player=turn) semantics."""
        G = self.G
        if _CORE is not None:
            if not hasattr(self, '_obs_scalar'):
                self._obs_scalar = np.empty((G, 18), dtype=np.float32)
                self._obs_planes = np.empty((G, 7, BOARD_N, BOARD_N),
                                            dtype=np.float32)
            _CORE.observations_core(self.board, self.piece_cnt, self.color,
                                    self._obs_scalar, self._obs_planes)
            return self._obs_scalar, self._obs_planes
        me = self.color
        b = self.board
        col = b >> 1
        occ = b >= 0
        planes = np.zeros((G, 7, N_CELLS), dtype=np.float32)
        planes[:, 0] = 1.0
        planes[:, 1] = occ & (col == me[:, None])
        planes[:, 2] = occ & (col == (me[:, None] ^ 1))
        planes[:, 3] = b == (me * 2)[:, None]
        planes[:, 4] = b == (me * 2 + 1)[:, None]
        # planes 5/6 (true opponent types) stay zero: partial view
        w = me == WHITE
        planes[w] = planes[w][:, :, ::-1]        # 180-degree rotation
        scalar = np.zeros((G, 18), dtype=np.float32)
        scalar[:, 0] = me == BLACK
        scalar[:, 1] = 1.0                        # turn view
        codes = np.stack([me * 2, me * 2 + 1, (me ^ 1) * 2, (me ^ 1) * 2 + 1],
                         axis=1)                  # (G, 4) viewer-relative
        n = np.take_along_axis(self.piece_cnt.astype(np.int64), codes, axis=1)
        for g in range(4):
            for i in range(4):
                scalar[:, 2 + 4 * g + i] = n[:, g] == i + 1
        return scalar, planes.reshape(G, 7, BOARD_N, BOARD_N)

    # -- transition -------------------------------------------------------
    def step(self, actions):
        """Apply one action per game (entries for finished games ignored;
        actions are assumed legal).  Returns the (G,) bool mask of games
        that finished this step."""
        act = np.asarray(actions, dtype=np.int64)
        if _CORE is not None:
            if not hasattr(self, '_done_buf'):
                self._done_buf = np.empty(self.G, dtype=bool)
            _CORE.step_core(self.board, self.slot_of, self.piece_pos,
                            self.piece_cnt, self.color, self.turn_count,
                            self.win, self.over, np.ascontiguousarray(act),
                            self._done_buf)
            return self._done_buf
        active = self.win < 0
        lay = active & (self.turn_count < 0)

        gl = np.nonzero(lay)[0]
        if len(gl):
            c = self.color[gl]
            blues = LAYOUT_BLUE[act[gl] - N_MOVE_ACTIONS]        # (n, 8)
            codes = (c[:, None] * 2 + (~blues)).astype(np.int8)
            cells = START_CELLS[c]                               # (n, 8)
            slots = (c[:, None] * 8 + np.arange(8)).astype(np.int8)
            self.board[gl[:, None], cells] = codes
            self.slot_of[gl[:, None], cells] = slots
            self.piece_pos[gl[:, None], slots] = cells.astype(np.int8)
            self.piece_cnt[gl, c * 2] = 4
            self.piece_cnt[gl, c * 2 + 1] = 4

        gm = np.nonzero(active & ~lay)[0]
        if len(gm):
            c = self.color[gm]
            a = act[gm]
            fcell = A_FROM[c, a]
            fcode = self.board[gm, fcell]
            fslot = self.slot_of[gm, fcell].astype(np.int64)
            tcell = A_TO[c, a]
            off = tcell < 0

            go = gm[off]                 # a blue walks off through the goal
            if len(go):
                self.board[go, fcell[off]] = -1
                self.slot_of[go, fcell[off]] = -1
                self.piece_pos[go, fslot[off]] = -1
                self.piece_cnt[go, fcode[off].astype(np.int64)] -= 1
                self.win[go] = c[off].astype(np.int8)

            gn = gm[~off]
            if len(gn):
                cn = c[~off]
                f2, t2 = fcell[~off], tcell[~off]
                fc2, fs2 = fcode[~off], fslot[~off]
                tcode = self.board[gn, t2]
                cap = tcode >= 0
                gc = gn[cap]
                if len(gc):
                    tcap = tcode[cap].astype(np.int64)
                    tslot = self.slot_of[gc, t2[cap]].astype(np.int64)
                    self.piece_pos[gc, tslot] = -1
                    self.piece_cnt[gc, tcap] -= 1
                    wiped = self.piece_cnt[gc, tcap] == 0
                    mover = cn[cap]
                    # all enemy blues captured: mover wins; capturing all
                    # enemy reds makes the CAPTURER lose
                    win_val = np.where((tcap & 1) == BLUE, mover, mover ^ 1)
                    self.win[gc[wiped]] = win_val[wiped].astype(np.int8)
                self.board[gn, f2] = -1
                self.slot_of[gn, f2] = -1
                self.board[gn, t2] = fc2
                self.slot_of[gn, t2] = fs2.astype(np.int8)
                self.piece_pos[gn, fs2] = t2.astype(np.int8)

        ga = np.nonzero(active)[0]
        self.color[ga] ^= 1
        self.turn_count[ga] += 1
        draw = active & (self.turn_count >= MAX_TURNS) & (self.win < 0)
        self.win[draw] = 2
        done = active & (self.win >= 0)
        self.over = self.win >= 0
        return done

    # -- outcome ----------------------------------------------------------
    def outcomes(self, games):
        """(n, 2) per-player outcomes for the given finished games."""
        w = self.win[games]
        oc = np.zeros((len(games), 2), dtype=np.float32)
        oc[w == BLACK] = (1.0, -1.0)
        oc[w == WHITE] = (-1.0, 1.0)
        return oc
