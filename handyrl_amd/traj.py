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

    def episode_indices(self, g_rows, lens):
        """Flat (sum(lens),) gather indices for finished episodes:
        (g repeated len(g) times, t = 0..len(g)-1).  Fully vectorized —
        the short-episode regime commits hundreds of episodes per
        service, so a per-episode python loop here costs ~20% of the
        whole actor phase."""
        lens = np.asarray(lens, dtype=np.int64)
        if lens.size == 0:
            empty = np.empty(0, dtype=np.int64)
            return empty, empty
        g_flat = np.repeat(np.asarray(g_rows, dtype=np.int64), lens)
        starts = np.repeat(np.cumsum(lens) - lens, lens)
        t_flat = np.arange(int(lens.sum()), dtype=np.int64) - starts
        return g_flat, t_flat
