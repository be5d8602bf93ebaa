"""Vectorized Hungry Geese engine: G games stepped at once in numpy.

Same rules as handyrl_amd.envs.hungry_geese.GeeseState (the single-game
oracle; parity is tested in tests/test_vec_geese.py), restructured as
struct-of-arrays so the GPU actor pool (handyrl_amd/actor.py) can run
hundreds of self-play games against ONE batched network forward per step —
the MI355X replacement for the reference's one-process-per-environment
workers (reference worker.py / generation.py).

Goose bodies are fixed-capacity ring buffers (head at ``start``, elements
at (start + i) % CAP), so head-push/tail-pop are O(1) index updates and
board grids are maintained incrementally.
"""

import os

import numpy as np

from .hungry_geese import (ROWS, COLS, N_CELLS, N_PLAYERS, HUNGER_RATE,
                           MIN_FOOD, MAX_LEN, MAX_STEPS, OPPOSITE)


def _load_native_core():
    """In-tree C++ env core (envs/src/vec_geese_core.cpp): the RNG-free
    step phases and observation build as native loops.  Bit-equal to the
    numpy engine (tests/test_vec_geese_native.py); HANDYRL_NO_NATIVE_ENV=1
    or a missing .so falls back to pure numpy."""
    if os.environ.get('HANDYRL_NO_NATIVE_ENV') == '1':
        return None
    so = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                      '_vec_geese_core.so')
    if not os.path.exists(so):
        return None
    try:
        import importlib.machinery
        import importlib.util
        loader = importlib.machinery.ExtensionFileLoader('_vec_geese_core', so)
        spec = importlib.util.spec_from_loader('_vec_geese_core', loader)
        mod = importlib.util.module_from_spec(spec)
        loader.exec_module(mod)
        return mod
    except Exception:            # noqa: BLE001 - numpy fallback
        return None


_CORE = _load_native_core()

CAP = N_CELLS  # a goose can never exceed the board

# SHIFT[cell, action] -> next cell on the torus
_r, _c = np.divmod(np.arange(N_CELLS), COLS)
SHIFT = np.stack([
    ((_r - 1) % ROWS) * COLS + _c,          # NORTH
    ((_r + 1) % ROWS) * COLS + _c,          # SOUTH
    _r * COLS + (_c - 1) % COLS,            # WEST
    _r * COLS + (_c + 1) % COLS,            # EAST
], axis=1).astype(np.int32)

_OPP = np.array(OPPOSITE, dtype=np.int32)

# seat-relative channel gather: obs channel c of player k shows goose (c+k)%4
_REL = (np.arange(N_PLAYERS)[None, :] + np.arange(N_PLAYERS)[:, None]) % N_PLAYERS

# CHMAP[k, c] = canonical channel feeding seat k's obs channel c.  The
# canonical per-game encoding is [heads 0-3, tails 4-7, bodies 8-11,
# prev-heads 12-15, food 16] in GOOSE order; a seat view is the same data
# with the four goose planes of each group rotated so the seat's own goose
# comes first.  Seat expansion is a pure channel gather — done on the GPU
# (ops.obs_to_nhwc_rot / an index_select in the replay gather), never on
# the host hot path.
_c = np.arange(17)
CHMAP = np.where(_c[None, :] < 16,
                 (_c[None, :] & 12) | ((_c[None, :] + np.arange(N_PLAYERS)[:, None]) & 3),
                 16).astype(np.int64)          # (4, 17)


class GeeseVecEnv:
    """G simultaneous Hungry Geese games."""

    def __init__(self, n_games, seed=0):
        self.G = n_games
        self.rng = np.random.default_rng(seed)
        G = self.G
        self.body = np.full((G, N_PLAYERS, CAP), -1, dtype=np.int32)
        self.start = np.zeros((G, N_PLAYERS), dtype=np.int32)
        self.length = np.zeros((G, N_PLAYERS), dtype=np.int32)
        self.alive = np.zeros((G, N_PLAYERS), dtype=bool)
        self.scores = np.zeros((G, N_PLAYERS), dtype=np.float64)
        self.last_action = np.full((G, N_PLAYERS), -1, dtype=np.int32)
        self.prev_head = np.full((G, N_PLAYERS), -1, dtype=np.int32)
        self.food = np.full((G, MIN_FOOD), -1, dtype=np.int32)
        self.step_count = np.zeros(G, dtype=np.int32)
        self.over = np.zeros(G, dtype=bool)
        # incremental grids
        self.body_grid = np.zeros((G, N_PLAYERS, N_CELLS), dtype=np.uint8)
        self._gp_arange = np.arange(G * N_PLAYERS)
        self.reset_games(np.arange(G))

    # -- helpers -----------------------------------------------------------
    def _head(self, gmask=None):
        flat = self.body.reshape(self.G * N_PLAYERS, CAP)
        return flat[self._gp_arange, (self.start % CAP).ravel()] \
            .reshape(self.G, N_PLAYERS)

    def _tail_cell(self):
        flat = self.body.reshape(self.G * N_PLAYERS, CAP)
        idx = ((self.start + self.length - 1) % CAP).ravel()
        return flat[self._gp_arange, idx].reshape(self.G, N_PLAYERS)

    def _tail_at(self, gi, pi):
        """Tail cells for the (gi, pi) subset only (avoids the full-grid
        gather on the hot path)."""
        idx = (self.start[gi, pi] + self.length[gi, pi] - 1) % CAP
        return self.body[gi, pi, idx]

    def reset_games(self, games):
        """Reset the given game indices to fresh initial states."""
        if len(games) == 0:
            return
        # distinct random cells per game: the K smallest of iid uniforms
        # (argpartition) = a uniform sample without replacement
        K = N_PLAYERS + MIN_FOOD
        cells = np.argpartition(self.rng.random((len(games), N_CELLS)),
                                K, axis=1)[:, :K].astype(np.int32)
        self.body[games] = -1
        self.body[games, :, 0] = cells[:, :N_PLAYERS]
        self.food[games] = cells[:, N_PLAYERS:]
        self.start[games] = 0
        self.length[games] = 1
        self.alive[games] = True
        self.scores[games] = 0.0
        self.last_action[games] = -1
        self.prev_head[games] = -1
        self.step_count[games] = 0
        self.over[games] = False
        self.body_grid[games] = 0
        gg = np.repeat(games, N_PLAYERS)
        pp = np.tile(np.arange(N_PLAYERS), len(games))
        self.body_grid[gg, pp, self.body[gg, pp, 0]] = 1

    def _kill(self, g_idx, p_idx):
        self.alive[g_idx, p_idx] = False
        self.length[g_idx, p_idx] = 0
        self.body_grid[g_idx, p_idx] = 0

    def step(self, actions):
        """actions: (G, 4) int32; entries for dead seats/finished games ignored."""
        G = self.G
        act = np.asarray(actions, dtype=np.int32)
        if _CORE is not None:
            _CORE.step_core(self.body, self.start, self.length, self.alive,
                            self.last_action, self.prev_head, self.food,
                            self.step_count, self.over, self.body_grid,
                            np.ascontiguousarray(act))
            return self._step_tail()
        live = self.alive & ~self.over[:, None]

        self.prev_head = np.where(self.alive, self._head(), -1)

        # 1) reverse-move deaths
        rev = live & (self.last_action >= 0) & (act == _OPP[np.clip(self.last_action, 0, 3)])
        gi, pi = np.nonzero(rev)
        self._kill(gi, pi)
        live = self.alive & ~self.over[:, None]
        self.last_action = np.where(live, act, self.last_action)

        # 2) move: new head cell
        heads = self._head()
        new_head = np.where(live, SHIFT[np.clip(heads, 0, N_CELLS - 1), np.clip(act, 0, 3)], heads)

        # food consumption (order-free: contested food implies a head
        # collision, and the losers die this step anyway)
        ate = np.zeros((G, N_PLAYERS), dtype=bool)
        for f in range(MIN_FOOD):
            hit = live & (new_head == self.food[:, f][:, None]) & (self.food[:, f][:, None] >= 0)
            ate |= hit
            self.food[hit.any(axis=1), f] = -1

        # pop tail unless the goose ate (bodies never self-overlap, so the
        # grid bit of the vacated cell can be cleared directly); the live
        # index set is reused for the push below
        gl, pl = np.nonzero(live)
        pop = ~ate[gl, pl]
        gi, pi = gl[pop], pl[pop]
        if len(gi):
            tail = self._tail_at(gi, pi)
            self.length[gi, pi] -= 1
            self.body_grid[gi, pi, tail] = 0

        # push new head; a head landing on this goose's own remaining body is
        # a self-collision the 0/1 grid can't count — flag it explicitly
        self_crash = np.zeros((G, N_PLAYERS), dtype=bool)
        if len(gl):
            nh = new_head[gl, pl]
            self_crash[gl, pl] = self.body_grid[gl, pl, nh] == 1
            new_start = (self.start[gl, pl] - 1) % CAP
            self.start[gl, pl] = new_start
            self.body[gl, pl, new_start] = nh
            self.length[gl, pl] += 1
            self.body_grid[gl, pl, nh] = 1

        # 3) hunger shrink every HUNGER_RATE transitions
        hungry_games = (self.step_count + 1) % HUNGER_RATE == 0
        shrink = live & hungry_games[:, None]
        gi, pi = np.nonzero(shrink)
        if len(gi):
            tail = self._tail_at(gi, pi)
            self.length[gi, pi] -= 1
            keep = self_crash[gi, pi] & (tail == new_head[gi, pi])  # looped head
            self.body_grid[gi[~keep], pi[~keep], tail[~keep]] = 0
            starved = self.length[gi, pi] <= 0
            self._kill(gi[starved], pi[starved])
        live = self.alive & ~self.over[:, None]

        # 4) collision deaths: live head on a cell with >1 segment (gather
        # the 4 per-player grid bits at each live head instead of reducing
        # the whole grid)
        heads = self._head()
        gi, pi = np.nonzero(live)
        if len(gi):
            seg = self.body_grid[gi, :, heads[gi, pi]]          # (n, 4)
            crash = (seg.sum(axis=1) > 1) | self_crash[gi, pi]
            self._kill(gi[crash], pi[crash])
        live = self.alive & ~self.over[:, None]

        return self._step_tail()

    def _step_tail(self):
        """Post-move phases shared by the numpy and native cores: food
        replenishment (the ONLY rng consumer in step — kept in python so
        both cores read the identical stream), step counting, scores and
        termination."""
        # 5) food replenishment onto random free cells (vectorized over the
        # needy games: argmax of iid uniforms over the free set = a uniform
        # free-cell draw, matching the oracle's shuffle-and-take semantics)
        need_mask = (self.food < 0) & ~self.over[:, None]
        ng = np.nonzero(need_mask.any(axis=1))[0]
        if len(ng):
            occ = self.body_grid[ng].any(axis=1)            # (K, cells)
            for f in range(MIN_FOOD):
                have = self.food[ng, f] >= 0
                occ[np.nonzero(have)[0], self.food[ng[have], f]] = True
            for f in range(MIN_FOOD):
                miss = self.food[ng, f] < 0
                if not miss.any():
                    continue
                idx = np.nonzero(miss)[0]
                r = self.rng.random((len(idx), N_CELLS))
                r[occ[idx]] = -1.0
                cell = r.argmax(axis=1)
                ok = r[np.arange(len(idx)), cell] >= 0      # any free cell
                self.food[ng[idx[ok]], f] = cell[ok].astype(np.int32)
                occ[idx[ok], cell[ok]] = True

        stepped = ~self.over
        self.step_count[stepped] += 1

        # 6) scores for survivors
        surv = self.alive & stepped[:, None]
        self.scores[surv] = (self.step_count[:, None] * (MAX_LEN + 1) + self.length)[surv]

        # 7) termination
        done = stepped & ((self.alive.sum(axis=1) <= 1) | (self.step_count >= MAX_STEPS))
        self.over |= done
        return done

    # -- observation / outcome ----------------------------------------------
    def observations(self):
        """uint8 (G, 17, 7, 11): CANONICAL per-game planes (goose order
        0-3 per group; see CHMAP).  Seat views are channel gathers done on
        the GPU; use observations_per_seat() for host-side parity checks."""
        G = self.G
        if not hasattr(self, '_obs_buf'):
            self._obs_buf = np.zeros((G, 17, N_CELLS), dtype=np.uint8)
        obs = self._obs_buf
        if _CORE is not None:
            _CORE.observations_core(self.body, self.start, self.length,
                                    self.alive, self.prev_head, self.food,
                                    self.body_grid, obs)
            return obs.reshape(G, 17, ROWS, COLS)
        head_grid = obs[:, 0:4]
        tail_grid = obs[:, 4:8]
        prev_grid = obs[:, 12:16]
        food_grid = obs[:, 16]
        head_grid[:] = 0
        tail_grid[:] = 0
        prev_grid[:] = 0
        food_grid[:] = 0
        heads = self._head()
        tails = self._tail_cell()
        gi, pi = np.nonzero(self.alive)
        head_grid[gi, pi, heads[gi, pi]] = 1
        tail_grid[gi, pi, tails[gi, pi]] = 1
        gi, pi = np.nonzero(self.prev_head >= 0)
        prev_grid[gi, pi, self.prev_head[gi, pi]] = 1
        obs[:, 8:12] = self.body_grid
        for f in range(MIN_FOOD):
            ok = self.food[:, f] >= 0
            food_grid[np.nonzero(ok)[0], self.food[ok, f]] = 1
        return obs.reshape(G, 17, ROWS, COLS)

    def observations_per_seat(self):
        """uint8 (G, 4, 17, 7, 11): seat-expanded views (CPU fallback /
        tests) — CHMAP channel gather of the canonical encoding."""
        canon = self.observations().reshape(self.G, 17, N_CELLS)
        return canon[:, CHMAP].reshape(self.G, N_PLAYERS, 17, ROWS, COLS)

    def outcomes(self, games):
        """Pairwise rank outcome per seat for the given finished games."""
        sc = self.scores[games]                        # (n, 4)
        gt = (sc[:, :, None] > sc[:, None, :]).sum(axis=2)
        lt = (sc[:, :, None] < sc[:, None, :]).sum(axis=2)
        return (gt - lt) / (N_PLAYERS - 1)
