"""Device-side trajectory recording for the GPU actor path.

Round-1 measurement showed the flagship pipeline host-bound: every
observation was recorded into host columnar buffers by the env workers,
packaged per episode, pickled over a pipe to the parent, staged into
pinned memory and H2D-copied into the device replay ring — even though
the SAME bytes had already crossed to the GPU once for the batched
inference forward.

This module closes that loop the MI355X way: the actor graph itself
scatters each service's observations (already on device), the computed
(action, prob, value) rows and the derived alive mask into per-game
trajectory rings in HBM3E (`TrajRecorder`, plain tensor index-writes —
hipGraph-capturable, so recording costs zero extra host work).  When an
episode finishes, only (game row, length, outcome) metadata crosses the
pipe; `DeviceReplay.commit_traj` then copies the finished rows
device-to-device into the replay ring at ~8 TB/s instead of re-uploading
them from the host.

Replaces the host half of reference generation.py:31-91 (moment
recording + packaging) for GPU actors.
"""

import numpy as np
import torch

from .envs.hungry_geese import MAX_STEPS

OBS_SHAPE = (17, 7, 11)


class TrajRecorder:
    """Per-game trajectory rings in device memory.

    Row ``g`` of each ring holds the in-progress episode of global game
    ``g`` (games are dense across workers/slots: worker w, slot s, local
    game i -> row base(w, s) + i).  Row ``n_games`` (the last row) is a
    scratch target for padded bucket rows, so captured graphs can always
    scatter a full bucket.
    """

    def __init__(self, n_games, device, max_steps=MAX_STEPS):
        self.n_games = n_games
        self.max_steps = max_steps
        self.device = device
        R = n_games + 1                           # +1 scratch row
        self.obs = torch.zeros((R, max_steps) + OBS_SHAPE, dtype=torch.uint8,
                               device=device)
        self.alive = torch.zeros((R, max_steps, 4), dtype=torch.bool,
                                 device=device)
        # packed (action, prob, value) per seat as produced by the actor
        # graph; action is cast to int32 at commit time
        self.rec = torch.zeros((R, max_steps, 4, 3), dtype=torch.float32,
                               device=device)
        self.scratch_row = n_games

    def record_(self, obs_u8, packed, gidx, tidx):
        """Scatter one service's rows into the rings (called INSIDE the
        captured actor forward; every op is graph-capturable).

        obs_u8: (B, 17, 7, 11) uint8 on device (canonical, 1 row/game)
        packed: (B*4, 3) float32 on device (action, prob, value per seat)
        gidx/tidx: (B,) int64 on device (global game row, step index)
        """
        B = obs_u8.shape[0]
        self.obs[gidx, tidx] = obs_u8
        # a seat is alive iff its head plane has a set cell
        alive = obs_u8[:, :4].reshape(B, 4, -1).amax(-1) > 0
        self.alive[gidx, tidx] = alive
        self.rec[gidx, tidx] = packed.reshape(B, 4, 3)

    @staticmethod
    def flat_indices(g_rows, lens):
        """Flat (sum(lens),) gather indices for finished episodes:
        (g repeated len(g) times, t = 0..len(g)-1).  Fully vectorized."""
        lens = np.asarray(lens, dtype=np.int64)
        if lens.size == 0:
            empty = np.empty(0, dtype=np.int64)
            return empty, empty
        g_flat = np.repeat(np.asarray(g_rows, dtype=np.int64), lens)
        starts = np.repeat(np.cumsum(lens) - lens, lens)
        t_flat = np.arange(int(lens.sum()), dtype=np.int64) - starts
        return g_flat, t_flat

    def episode_indices(self, g_rows, lens):
        """See flat_indices (kept as a method for the replay commit)."""
        return self.flat_indices(g_rows, lens)


class GeisterTrajRecorder:
    """Turn-based device trajectory rings for the Geister DRC actors.

    Row g holds the in-progress episode of global game g (the MOVER's
    view per step, matching the columnar turn-based episode format);
    row ``n_games`` is the scratch target for padded rows.  record_ runs
    INSIDE the engine's captured graph."""

    MAX_STEPS = 202                       # 2 layout turns + 200 moves

    def __init__(self, n_games, device, max_steps=MAX_STEPS):
        self.n_games = n_games
        self.max_steps = max_steps
        self.device = device
        R = n_games + 1
        self.scalar = torch.zeros(R, max_steps, 18, dtype=torch.uint8,
                                  device=device)
        self.board = torch.zeros(R, max_steps, 7, 6, 6, dtype=torch.uint8,
                                 device=device)
        self.mask = torch.zeros(R, max_steps, 214, dtype=torch.bool,
                                device=device)
        self.turn = torch.zeros(R, max_steps, dtype=torch.int8,
                                device=device)
        # (action, prob, value) as produced by the engine's packed output
        self.apv = torch.zeros(R, max_steps, 3, dtype=torch.float32,
                               device=device)
        self.scratch_row = n_games

    def record_(self, scalar_f, board_f, mask_f, parity, packed, gidx, tidx):
        """Scatter one engine service's rows (B = shard games) into the
        rings; every op is graph-capturable.  mask_f is the additive
        float mask (0 legal / 1e32 illegal)."""
        self.scalar[gidx, tidx] = scalar_f.to(torch.uint8)
        self.board[gidx, tidx] = board_f.to(torch.uint8)
        self.mask[gidx, tidx] = mask_f == 0.0
        self.turn[gidx, tidx] = parity.to(torch.int8)
        self.apv[gidx, tidx] = packed[:, :3]

    episode_indices = TrajRecorder.episode_indices
    flat_indices = staticmethod(TrajRecorder.flat_indices)
