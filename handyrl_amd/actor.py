"""GPU actor pool: batched self-play over vectorized environments.

The MI355X replacement for the reference's one-CPU-process-per-environment
workers (reference worker.py:26-89, generation.py:20-93): hundreds of
Hungry Geese games advance in lock-step; every step does ONE batched
network forward for all alive seats (bf16, inference mode) and one fused
masked-softmax-sample kernel (handyrl_amd/ops), instead of per-env
single-sample CPU inference.

Finished games are packaged into reference-format episodes (uncompressed
moment blocks by default — same interchange as the CPU worker path, minus
the bz2 that a local learner does not need).
"""

import os

import numpy as np
import torch

# avoid per-shape MIOpen exhaustive search; find-db lookups only
os.environ.setdefault('MIOPEN_FIND_MODE', 'FAST')

from . import ops
from .envs.vec_geese import GeeseVecEnv, N_PLAYERS
from .envs.hungry_geese import MAX_STEPS

MOMENT_KEYS = ('observation', 'selected_prob', 'action_mask', 'action',
               'value', 'reward', 'return')


class GeeseActorPool:
    """Self-play actor pool for Hungry Geese on one GPU."""

    def __init__(self, model, args, n_games=256, device=None, seed=0,
                 store_uint8_obs=True, use_graphs=True):
        self.args = args
        self.device = device if device is not None else (
            torch.device('cuda') if torch.cuda.is_available() else torch.device('cpu'))
        self.model = model
        self.vec = GeeseVecEnv(n_games, seed=seed)
        self.n_games = n_games
        self.store_uint8_obs = store_uint8_obs
        self.gamma = args.get('gamma', 0.8)
        self.compress = args.get('compress_episodes', False)
        self.compress_steps = args.get('compress_steps', 4)
        # columnar trajectory recording: struct-of-arrays ring per game
        # (episodes stay columnar through the replay buffer and batch maker)
        G, CAPT = n_games, MAX_STEPS
        self.rec_obs = np.zeros((G, CAPT, N_PLAYERS, 17, 7, 11), dtype=np.uint8)
        self.rec_alive = np.zeros((G, CAPT, N_PLAYERS), dtype=bool)
        self.rec_act = np.zeros((G, CAPT, N_PLAYERS), dtype=np.int32)
        self.rec_prob = np.zeros((G, CAPT, N_PLAYERS), dtype=np.float32)
        self.rec_val = np.zeros((G, CAPT, N_PLAYERS), dtype=np.float32)
        self.rec_len = np.zeros(G, dtype=np.int32)
        self.completed = []
        self.frames = 0          # env transitions executed (sum over games)
        self.episodes_done = 0
        self._zero_mask = None
        self.graphed = None
        self.fused = None
        if self.device.type == 'cuda' and os.environ.get('HANDYRL_NO_FUSED') != '1' \
                and hasattr(model, 'stem') and ops.available():
            from .models.geese_net import GeeseFusedEval
            self.fused = GeeseFusedEval(model, self.device)
        if use_graphs and self.device.type == 'cuda':
            from .hipgraph import GraphedActorForward
            self.graphed = GraphedActorForward(model, self.device, fused=self.fused)
        self.timing = {'obs': 0.0, 'fwd': 0.0, 'sample': 0.0,
                       'record': 0.0, 'env': 0.0, 'package': 0.0, 'n': 0}

    @torch.inference_mode()
    def _policy_forward(self, obs_f):
        if self.device.type == 'cuda':
            with torch.autocast('cuda', dtype=torch.bfloat16):
                out = self.model(obs_f, None)
            return out['policy'].float(), out['value'].float()
        out = self.model(obs_f, None)
        return out['policy'].float(), out['value'].float()

    def step_once(self):
        """Advance every live game by one transition; returns #frames."""
        import time
        tm = self.timing
        vec = self.vec
        t0 = time.time()
        obs_u8 = vec.observations()                     # (G, 4, 17, 7, 11)
        tm['obs'] += time.time() - t0
        live = vec.alive & ~vec.over[:, None]
        gi, pi = np.nonzero(live)
        if len(gi) == 0:
            return 0
        t0 = time.time()

        obs_sel = obs_u8[gi, pi]                        # (M, 17, 7, 11)
        M = len(gi)
        if self.graphed is not None:
            # hipGraph path: one static-buffer H2D copy + one graph replay
            # (forward + sample fused) + one packed D2H readback
            packed = self.graphed.run(torch.from_numpy(obs_sel)).cpu().numpy()
            tm['fwd'] += time.time() - t0
            t0 = time.time()
            actions = packed[:, 0].astype(np.int64)
            probs = packed[:, 1]
            values = packed[:, 2]
        elif self.device.type == 'cuda':
            # eager GPU path: pad to a fixed bucket so MIOpen keeps one
            # solution per shape
            bucket = 256 * ((M + 255) // 256)
            if bucket > M:
                pad = np.zeros((bucket - M,) + obs_sel.shape[1:], dtype=obs_sel.dtype)
                obs_in = np.concatenate([obs_sel, pad], axis=0)
            else:
                obs_in = obs_sel
            obs_t = torch.from_numpy(obs_in).to(self.device, non_blocking=True)
            policy, value = self._policy_forward(obs_t.float())
            policy, value = policy[:M], value[:M]
            A = policy.shape[1]
            tm['fwd'] += time.time() - t0
            t0 = time.time()
            if self._zero_mask is None or self._zero_mask.shape[0] < M:
                self._zero_mask = torch.zeros(max(M, 1), A, device=self.device)
            uniform = torch.rand(M, device=self.device)
            actions_t, probs_t = ops.masked_sample(policy, self._zero_mask[:M], uniform)
            # ONE device-to-host transfer for the step's three result vectors
            packed = torch.cat([actions_t.float().unsqueeze(1),
                                probs_t.unsqueeze(1), value], dim=1).cpu().numpy()
            actions = packed[:, 0].astype(np.int64)
            probs = packed[:, 1]
            values = packed[:, 2]
        else:
            # CPU path (tests / debugging)
            policy, value = self._policy_forward(torch.from_numpy(obs_sel).float())
            tm['fwd'] += time.time() - t0
            t0 = time.time()
            probs_full = torch.softmax(policy, dim=-1)
            actions_t = torch.multinomial(probs_full, 1).squeeze(-1)
            actions = actions_t.numpy()
            probs = probs_full.gather(-1, actions_t.unsqueeze(-1)).squeeze(-1).numpy()
            values = value.squeeze(-1).numpy()

        tm['sample'] += time.time() - t0
        t0 = time.time()
        act_grid = np.zeros((self.n_games, N_PLAYERS), dtype=np.int32)
        act_grid[gi, pi] = actions

        prob_row = np.zeros((self.n_games, N_PLAYERS), dtype=np.float32)
        val_row = np.zeros((self.n_games, N_PLAYERS), dtype=np.float32)
        prob_row[gi, pi] = probs
        val_row[gi, pi] = values

        # columnar recording: one fancy-indexed scatter per field
        game_has_live = live.any(axis=1)
        lg = np.nonzero(game_has_live)[0]
        t_idx = self.rec_len[lg]
        self.rec_obs[lg, t_idx] = obs_u8[lg]
        self.rec_alive[lg, t_idx] = live[lg]
        self.rec_act[lg, t_idx] = act_grid[lg]
        self.rec_prob[lg, t_idx] = prob_row[lg]
        self.rec_val[lg, t_idx] = val_row[lg]
        self.rec_len[lg] += 1

        tm['record'] += time.time() - t0
        t0 = time.time()
        done = vec.step(act_grid)
        tm['env'] += time.time() - t0
        t0 = time.time()
        self.frames += int(game_has_live.sum())

        finished = np.nonzero(done)[0]
        if len(finished):
            outcomes = vec.outcomes(finished)
            for k, g in enumerate(finished):
                self.completed.append(self._package(g, outcomes[k]))
                self.rec_len[g] = 0
            self.episodes_done += len(finished)
            vec.reset_games(finished)
        tm['package'] += time.time() - t0
        tm['n'] += 1
        return int(game_has_live.sum())

    def _package(self, g, outcome_row):
        """Slice the game's columnar recording into an episode dict."""
        S = int(self.rec_len[g])
        job_args = {'player': list(range(N_PLAYERS)),
                    'model_id': {p: -1 for p in range(N_PLAYERS)}}
        return {
            'args': job_args,
            'steps': S,
            'outcome': {p: float(outcome_row[p]) for p in range(N_PLAYERS)},
            'columnar': True,
            'n_actions': 4,
            'obs': self.rec_obs[g, :S].copy(),
            'alive': self.rec_alive[g, :S].copy(),
            'action': self.rec_act[g, :S].copy(),
            'prob': self.rec_prob[g, :S].copy(),
            'value': self.rec_val[g, :S].copy(),
        }

    def refresh_weights(self):
        """Re-fold BN into the packed MFMA weights after an optimizer step
        (the hand-written inference path then runs <=1 step stale)."""
        if self.fused is not None:
            self.fused.refresh()

    def harvest(self):
        """Return and clear the finished-episode list."""
        out = self.completed
        self.completed = []
        return out
