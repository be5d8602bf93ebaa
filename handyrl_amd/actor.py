"""GPU actor pool: batched self-play over vectorized environments.

The MI355X replacement for the reference's one-CPU-process-per-environment
workers (reference worker.py:26-89, generation.py:20-93): hundreds of
Hungry Geese games advance in lock-step; every step does ONE batched
network forward for all alive seats (bf16, inference mode) and one fused
masked-softmax-sample kernel (handyrl_amd/ops), instead of per-env
single-sample CPU inference.

Finished games are packaged as COLUMNAR episodes (struct-of-arrays) —
the replay buffer and batch maker accept those alongside the reference
moment-dict format used by remote CPU workers.
"""

import os
import sys

import numpy as np
import torch

# avoid per-shape MIOpen exhaustive search; find-db lookups only
os.environ.setdefault('MIOPEN_FIND_MODE', 'FAST')

from . import ops
from .envs.vec_geese import GeeseVecEnv, N_PLAYERS, CHMAP
from .envs.hungry_geese import MAX_STEPS

class GeeseActorPool:
    """Self-play actor pool for Hungry Geese on one GPU."""

    def __init__(self, model, args, n_games=256, device=None, seed=0,
                 use_graphs=True, engine=None, block_episodes=False,
                 record_host=True):
        # block_episodes: package ALL games finishing in a step as ONE
        # concatenated columnar block (env-worker processes; the parent's
        # drain thread splits it back into zero-copy per-episode views via
        # split_episode_block) — replaces per-episode dict building and
        # slice copies on the worker hot loop
        # record_host=False: device-side trajectory mode (handyrl_amd/traj):
        # the GPU records obs/alive/action/prob/value in HBM rings inside
        # the actor graph; this pool only tracks per-game step counters and
        # reports finished-episode metadata (take_meta / take_finished)
        self.block_episodes = block_episodes
        self.record_host = record_host
        self.args = args
        self.device = device if device is not None else (
            torch.device('cuda') if torch.cuda.is_available() else torch.device('cpu'))
        self.model = model
        self.vec = GeeseVecEnv(n_games, seed=seed)
        self.n_games = n_games
        # columnar trajectory recording: struct-of-arrays ring per game
        # (episodes stay columnar through the replay buffer and batch maker)
        G, CAPT = n_games, MAX_STEPS
        # obs are stored CANONICAL (per game, not per seat, 4x smaller);
        # the seat channel-gather happens on the GPU (CHMAP)
        if record_host:
            self.rec_obs = np.zeros((G, CAPT, 17, 7, 11), dtype=np.uint8)
            self.rec_alive = np.zeros((G, CAPT, N_PLAYERS), dtype=bool)
            self.rec_act = np.zeros((G, CAPT, N_PLAYERS), dtype=np.int32)
            self.rec_prob = np.zeros((G, CAPT, N_PLAYERS), dtype=np.float32)
            self.rec_val = np.zeros((G, CAPT, N_PLAYERS), dtype=np.float32)
        self.rec_len = np.zeros(G, dtype=np.int32)
        self._meta = None              # (rows, t_idx) of the last prepare
        self._finished = None          # (g_local, lens, outcomes) backlog
        self.completed = []
        self.frames = 0          # env transitions executed (sum over games)
        self.episodes_done = 0
        self._zero_mask = None
        self.graphed = None
        self.fused = None
        if engine is not None:
            self.fused, self.graphed = engine
        elif model is not None:
            if self.device.type == 'cuda' and os.environ.get('HANDYRL_NO_FUSED') != '1' \
                    and hasattr(model, 'conv0') and ops.available():
                from .models.geese_net import GeeseFusedEval
                self.fused = GeeseFusedEval(model, self.device)
            if use_graphs and self.device.type == 'cuda':
                from .hipgraph import GraphedActorForward
                self.graphed = GraphedActorForward(model, self.device, fused=self.fused)
        # pipelined-transfer staging (graphed path): obs are canonical
        # (one row per game), results are per-seat (4 rows per game)
        if self.graphed is not None:
            self._obs_pin = torch.empty(n_games, 17, 7, 11, dtype=torch.uint8,
                                        pin_memory=True)
            self._obs_pin_np = self._obs_pin.numpy()
            self._out_pin = torch.empty(N_PLAYERS * n_games, 3,
                                        dtype=torch.float32, pin_memory=True)
            self._out_pin_np = self._out_pin.numpy()
            self._event = torch.cuda.Event()
        self._pending = None
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
        self._phase1()
        return self._phase2()

    # -- external-inference mode (env-worker processes) --------------------
    def prepare_step(self, out_buf):
        """Build CANONICAL observations, record obs/alive columns, and
        write the live games' obs into ``out_buf[:M]`` (M = live games).
        The caller supplies per-seat (actions, probs, values) of length
        4*M in game-major seat order to complete_step."""
        import time
        tm = self.timing
        vec = self.vec
        t0 = time.time()
        obs_u8 = vec.observations()          # (G, 17, 7, 11) canonical
        tm['obs'] += time.time() - t0
        live = vec.alive & ~vec.over[:, None]
        lg = np.nonzero(live.any(axis=1))[0]
        if len(lg) == 0:
            self._pending = None
            self._meta = (lg.astype(np.int64), np.empty(0, dtype=np.int64))
            return 0
        t_idx = self.rec_len[lg]
        if self.record_host:
            self.rec_obs[lg, t_idx] = obs_u8[lg]
            self.rec_alive[lg, t_idx] = live[lg]
        else:
            self._meta = (lg.astype(np.int64), t_idx.astype(np.int64))
        self._rec_slot = (lg, t_idx)
        M = len(lg)
        out_buf[:M] = obs_u8[lg]
        self._pending = ('ext', lg, live, M)
        return M

    def take_meta(self):
        """(game rows, step indices) of the last prepare_step (traj mode)."""
        meta, self._meta = self._meta, None
        return meta

    def take_finished(self):
        """Finished-episode metadata accumulated since the last call
        (traj mode): (g_local int64[K], lens int64[K], outcomes f32[K,4])
        or None."""
        fin, self._finished = self._finished, None
        return fin

    def complete_step(self, actions, probs, values):
        """Apply externally computed inference results (4*M game-major
        seat rows; dead seats' rows are ignored); returns #frames."""
        assert self._pending is not None and self._pending[0] == 'ext'
        _, lg, live, M = self._pending
        self._pending = ('done', lg, live, M, actions, probs, values)
        return self._phase2()

    def _phase1(self):
        """Build observations and ISSUE the GPU forward (async on the
        graphed path: pinned H2D + graph replay + pinned D2H + event)."""
        import time
        tm = self.timing
        vec = self.vec
        t0 = time.time()
        obs_u8 = vec.observations()                     # (G, 17, 7, 11)
        tm['obs'] += time.time() - t0
        live = vec.alive & ~vec.over[:, None]
        lg = np.nonzero(live.any(axis=1))[0]
        if len(lg) == 0:
            self._pending = None
            return
        t0 = time.time()
        M = len(lg)

        # record the observation/alive columns now (obs is in hand and this
        # CPU work overlaps the GPU forward on the pipelined path)
        t_idx = self.rec_len[lg]
        self.rec_obs[lg, t_idx] = obs_u8[lg]
        self.rec_alive[lg, t_idx] = live[lg]
        self._rec_slot = (lg, t_idx)

        if self.graphed is not None:
            np.copyto(self._obs_pin_np[:M], obs_u8[lg])
            self.graphed.run_async(self._obs_pin, M, self._out_pin, self._event)
            self._pending = ('async', lg, live, M)
            tm['fwd'] += time.time() - t0
            return

        # eager fallback: expand canonical -> per-seat on host (tests)
        obs_sel = obs_u8.reshape(self.n_games, 17, 77)[lg][:, CHMAP] \
            .reshape(M * N_PLAYERS, 17, 7, 11)
        if self.device.type == 'cuda':
            # pad to a fixed bucket so MIOpen keeps one solution per shape
            R = M * N_PLAYERS
            bucket = 256 * ((R + 255) // 256)
            if bucket > R:
                pad = np.zeros((bucket - R,) + obs_sel.shape[1:], dtype=obs_sel.dtype)
                obs_in = np.concatenate([obs_sel, pad], axis=0)
            else:
                obs_in = obs_sel
            obs_t = torch.from_numpy(obs_in).to(self.device)
            policy, value = self._policy_forward(obs_t.float())
            policy, value = policy[:R], value[:R]
            A = policy.shape[1]
            if self._zero_mask is None or self._zero_mask.shape[0] < R:
                self._zero_mask = torch.zeros(max(R, 1), A, device=self.device)
            uniform = torch.rand(R, device=self.device)
            actions_t, probs_t = ops.masked_sample(policy, self._zero_mask[:R], uniform)
            packed = torch.cat([actions_t.float().unsqueeze(1),
                                probs_t.unsqueeze(1), value], dim=1).cpu().numpy()
            actions = packed[:, 0].astype(np.int64)
            probs = packed[:, 1]
            values = packed[:, 2]
        else:
            # CPU path (tests / debugging)
            policy, value = self._policy_forward(torch.from_numpy(obs_sel).float())
            probs_full = torch.softmax(policy, dim=-1)
            actions_t = torch.multinomial(probs_full, 1).squeeze(-1)
            actions = actions_t.numpy()
            probs = probs_full.gather(-1, actions_t.unsqueeze(-1)).squeeze(-1).numpy()
            values = value.squeeze(-1).numpy()
        tm['fwd'] += time.time() - t0
        self._pending = ('done', lg, live, M, actions, probs, values)

    def _phase2(self):
        """Wait for the issued forward, then apply actions: record, step
        the vectorized env, package finished games.  Returns #frames.
        Inference rows are game-major per-seat: row m*4+p is seat p of
        live game lg[m]; dead seats' rows carry garbage and are masked."""
        if self._pending is None:
            return 0
        import time
        tm = self.timing
        vec = self.vec
        pend, self._pending = self._pending, None
        t0 = time.time()
        if pend[0] == 'async':
            _, lg, live, M = pend
            self._event.synchronize()
            packed = self._out_pin_np[:M * N_PLAYERS]
            actions = packed[:, 0].astype(np.int64)
            probs = packed[:, 1]
            values = packed[:, 2]
        else:
            _, lg, live, M, actions, probs, values = pend
        tm['sample'] += time.time() - t0
        t0 = time.time()
        live_lg = live[lg]                               # (M, 4)
        act_grid = np.zeros((self.n_games, N_PLAYERS), dtype=np.int32)
        act_grid[lg] = np.where(live_lg, actions.reshape(M, N_PLAYERS), 0)

        # finish the columnar step record started in _phase1 (host mode;
        # in traj mode the GPU already recorded everything in-graph)
        lg, t_idx = self._rec_slot
        if self.record_host:
            prob_row = np.zeros((self.n_games, N_PLAYERS), dtype=np.float32)
            val_row = np.zeros((self.n_games, N_PLAYERS), dtype=np.float32)
            prob_row[lg] = np.where(live_lg, probs.reshape(M, N_PLAYERS), 0.0)
            val_row[lg] = np.where(live_lg, values.reshape(M, N_PLAYERS), 0.0)
            self.rec_act[lg, t_idx] = act_grid[lg]
            self.rec_prob[lg, t_idx] = prob_row[lg]
            self.rec_val[lg, t_idx] = val_row[lg]
        self.rec_len[lg] += 1

        tm['record'] += time.time() - t0
        t0 = time.time()
        done = vec.step(act_grid)
        tm['env'] += time.time() - t0
        t0 = time.time()
        self.frames += len(lg)

        finished = np.nonzero(done)[0]
        if len(finished):
            outcomes = vec.outcomes(finished)
            if not self.record_host:
                lens = self.rec_len[finished].astype(np.int64)
                item = (finished.astype(np.int64), lens,
                        outcomes.astype(np.float32))
                if self._finished is None:
                    self._finished = item
                else:                  # merge backlog (rare)
                    pf, pl, po = self._finished
                    self._finished = (np.concatenate([pf, item[0]]),
                                      np.concatenate([pl, item[1]]),
                                      np.concatenate([po, item[2]]))
                self.rec_len[finished] = 0
            elif self.block_episodes:
                self.completed.append(self._package_block(finished, outcomes))
                self.rec_len[finished] = 0
            else:
                for k, g in enumerate(finished):
                    self.completed.append(self._package(g, outcomes[k]))
                    self.rec_len[g] = 0
            self.episodes_done += len(finished)
            vec.reset_games(finished)
        tm['package'] += time.time() - t0
        tm['n'] += 1
        return len(lg)

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
            'canonical_obs': True,     # (S, 17, 7, 11); seat views = CHMAP gather
            'n_actions': 4,
            'obs': self.rec_obs[g, :S].copy(),
            'alive': self.rec_alive[g, :S].copy(),
            'action': self.rec_act[g, :S].copy(),
            'prob': self.rec_prob[g, :S].copy(),
            'value': self.rec_val[g, :S].copy(),
        }

    def _package_block(self, finished, outcome_rows):
        """One concatenated columnar block for every game finishing this
        step (worker hot path: 5 array concats + one dict instead of
        per-episode copies and dicts)."""
        lens = self.rec_len[finished].astype(np.int32)
        pairs = [(int(g), int(s)) for g, s in zip(finished, lens)]
        cat = lambda buf: np.concatenate([buf[g, :s] for g, s in pairs])
        return {
            'block': True, 'columnar': True, 'canonical_obs': True,
            'n_actions': 4, 'lens': lens,
            'outcome': outcome_rows.astype(np.float32),
            'obs': cat(self.rec_obs), 'alive': cat(self.rec_alive),
            'action': cat(self.rec_act), 'prob': cat(self.rec_prob),
            'value': cat(self.rec_val),
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


class PipelinedGeesePool:
    """Two actor shards in software pipeline: while shard A's forward runs
    on the GPU (issued async through pinned staging), shard B's CPU side
    (action unpack, columnar record, vectorized env step, episode
    packaging) executes — hiding one side under the other.  One shared
    fused-MFMA/hipGraph engine serves both shards."""

    def __init__(self, model, args, n_games=512, device=None, seed=0):
        half = max(1, n_games // 2)
        first = GeeseActorPool(model, args, n_games=half, device=device, seed=seed)
        engine = (first.fused, first.graphed)
        second = GeeseActorPool(model, args, n_games=n_games - half,
                                device=device, seed=seed + 7777, engine=engine)
        self.pools = [first, second]
        self.cur = 0
        self.calls_per_vec_step = 2     # one call completes half the games

    def step_once(self):
        """Issue shard A's forward; complete shard B's step. Returns the
        frames finished this call (~n_games/2)."""
        self.pools[self.cur]._phase1()
        frames = self.pools[1 - self.cur]._phase2()
        self.cur ^= 1
        return frames

    def drain(self):
        """Complete any in-flight shard step (call before harvesting all)."""
        total = 0
        for pool in self.pools:
            total += pool._phase2()
        return total

    def refresh_weights(self):
        self.pools[0].refresh_weights()   # the engine is shared

    def harvest(self):
        out = []
        for pool in self.pools:
            out.extend(pool.harvest())
        return out

    @property
    def episodes_done(self):
        return sum(p.episodes_done for p in self.pools)

    @property
    def frames(self):
        return sum(p.frames for p in self.pools)

    @property
    def timing(self):
        merged = {k: 0.0 for k in self.pools[0].timing}
        for p in self.pools:
            for k, v in p.timing.items():
                merged[k] += v
        merged['n'] = max(1, merged['n'] // 2)
        return merged


def split_episode_block(item):
    """Expand a concatenated episode block (GeeseActorPool._package_block)
    into per-episode dicts whose arrays are VIEWS into the block — the
    standard columnar episode format, no copies.  Non-block items pass
    through unchanged."""
    if not isinstance(item, dict) or not item.get('block'):
        return [item]
    job_args = {'player': list(range(N_PLAYERS)),
                'model_id': {p: -1 for p in range(N_PLAYERS)}}
    eps, off = [], 0
    oc = item['outcome']
    for i, S in enumerate(item['lens']):
        S = int(S)
        sl = slice(off, off + S)
        off += S
        eps.append({
            'args': job_args, 'steps': S,
            'outcome': {p: float(oc[i, p]) for p in range(N_PLAYERS)},
            'columnar': True, 'canonical_obs': item['canonical_obs'],
            'n_actions': item['n_actions'],
            'obs': item['obs'][sl], 'alive': item['alive'][sl],
            'action': item['action'][sl], 'prob': item['prob'][sl],
            'value': item['value'][sl],
        })
    return eps


def _geese_env_worker(conn, ep_conn, obs_name, res_name, n_games, args, seed,
                      slots=2, traj_mode=False):
    """Env-side child process: vectorized stepping (plus columnar
    recording and episode packaging in host-record mode) on host cores;
    observations/results move through shared memory, inference runs in
    the parent (GPU).

    The shard is split into ``slots`` software-pipelined halves: while one
    half's observations are away at the parent (GPU forward + service
    latency), the worker steps/records the other half — the inference
    round trip hides entirely behind env CPU work.

    With ``traj_mode`` the GPU records trajectories in HBM rings inside
    the actor graph (handyrl_amd/traj): this worker records NOTHING —
    each obs message carries the (game row, step index) metadata for the
    in-graph scatter plus finished-episode (rows, lens, outcomes) info,
    and the episode pipe goes unused.
    """
    from multiprocessing import shared_memory
    obs_shm = shared_memory.SharedMemory(name=obs_name)
    res_shm = shared_memory.SharedMemory(name=res_name)
    per = max(1, n_games // slots)
    obs_views, res_views, pools = [], [], []
    for s in range(slots):
        obs_views.append(np.ndarray((per, 17, 7, 11), dtype=np.uint8,
                                    buffer=obs_shm.buf,
                                    offset=s * per * 17 * 7 * 11))
        res_views.append(np.ndarray((per * N_PLAYERS, 3), dtype=np.float32,
                                    buffer=res_shm.buf,
                                    offset=s * per * N_PLAYERS * 12))
        pools.append(GeeseActorPool(
            None, args, n_games=per, device=torch.device('cpu'),
            use_graphs=False, seed=seed + 131 * s,
            record_host=not traj_mode,
            block_episodes=os.environ.get('HANDYRL_BLOCK_EPISODES',
                                          '1') == '1'))

    def obs_msg(s, m, frames):
        if traj_mode:
            # no per-step meta: shards are always full (asserted below)
            # and the parent mirrors the step counters itself
            pools[s].take_meta()
            return ('obs', s, m, frames, pools[s].take_finished())
        return ('obs', s, m, frames)

    def prep(s):
        m = pools[s].prepare_step(obs_views[s])
        if traj_mode and m != per:
            # the parent's whole-service graphs assume full shards (every
            # game alive until reset, which _phase2 does immediately)
            raise RuntimeError('traj mode shard not full: %d != %d'
                               % (m, per))
        return m

    m_inflight = [0] * slots
    for s in range(slots):                    # prime the pipeline
        m_inflight[s] = prep(s)
        conn.send(obs_msg(s, m_inflight[s], 0))
    while True:
        cmd = conn.recv()
        if cmd == 'quit':
            break
        s = cmd[1]                            # ('go', slot)
        M, frames = m_inflight[s], 0
        if M:
            r = res_views[s][:M * N_PLAYERS]      # 4 seat rows per game
            if traj_mode:                  # probs/values recorded on device
                frames = pools[s].complete_step(
                    r[:, 0].astype(np.int64), None, None)
            else:
                frames = pools[s].complete_step(
                    r[:, 0].astype(np.int64), r[:, 1].copy(), r[:, 2].copy())
            eps = pools[s].harvest()
            if eps:
                # episodes travel on their own pipe, drained by a parent
                # background thread: the service path never deserializes them
                ep_conn.send(eps)
        m_inflight[s] = prep(s)
        conn.send(obs_msg(s, m_inflight[s], frames))


class MultiProcGeesePool:
    """Actor pool with env work in W child processes: each child owns a
    game shard (vectorized stepping + columnar recording + packaging on
    its own core); the parent services children round-robin with the
    shared hipGraph/MFMA inference engine, one child's GPU batch in flight
    while the next child's staging overlaps.

    Construction forks the children — call it BEFORE any HIP context
    exists in the parent; attach(model, device) wires the engine after.
    """

    def __init__(self, args, n_games=768, seed=0, workers=3, slots=None,
                 traj_mode=False, make_stubs=True):
        import multiprocessing as mp
        from multiprocessing import shared_memory
        self.args = args
        self.workers = workers
        # traj_mode: device-side trajectory recording (handyrl_amd/traj) —
        # requires attach(model, device, replay=DeviceReplay) on CUDA;
        # workers then ship only step/episode METADATA, never obs arrays
        self.traj_mode = traj_mode
        # make_stubs=False skips per-episode stat stubs on the service
        # path (consumers that only need episodes_done, e.g. bench.py)
        self.make_stubs = make_stubs
        if slots is None:
            # >1 slots (double-buffered half-shards) measures faster but a
            # GPU-side transport race poisons recorded values (NaN losses
            # after ~100 learner steps, see BASELINE.md learning-sanity
            # section) — off until the race is found; opt in to reproduce.
            slots = int(os.environ.get('HANDYRL_ACTOR_SLOTS', '1'))
        per = max(1, n_games // workers)
        slots = max(1, min(slots, per))
        self.slots = slots
        self.calls_per_vec_step = workers * slots
        self.n_per = max(1, per // slots)       # games per slot
        self.conns, self.ep_conns, self.procs, self.shms = [], [], [], []
        self.obs_views, self.res_views = [], []  # [worker][slot]
        capg = self.n_per                        # canonical obs: 1 row/game
        capr = self.n_per * N_PLAYERS            # results: 4 seat rows/game
        for w in range(workers):
            obs_shm = shared_memory.SharedMemory(
                create=True, size=slots * capg * 17 * 7 * 11)
            res_shm = shared_memory.SharedMemory(
                create=True, size=slots * capr * 3 * 4)
            self.shms += [obs_shm, res_shm]
            self.obs_views.append([
                np.ndarray((capg, 17, 7, 11), dtype=np.uint8,
                           buffer=obs_shm.buf, offset=s * capg * 17 * 7 * 11)
                for s in range(slots)])
            self.res_views.append([
                np.ndarray((capr, 3), dtype=np.float32,
                           buffer=res_shm.buf, offset=s * capr * 12)
                for s in range(slots)])
            parent_conn, child_conn = mp.Pipe(duplex=True)
            ep_parent, ep_child = mp.Pipe(duplex=False)
            proc = mp.Process(target=_geese_env_worker,
                              args=(child_conn, ep_child, obs_shm.name,
                                    res_shm.name, per, args, seed + 977 * w,
                                    slots, traj_mode),
                              daemon=True)
            proc.start()
            child_conn.close()
            ep_child.close()
            self.conns.append(parent_conn)
            self.ep_conns.append(ep_parent)
            self.procs.append(proc)

        self.device = None
        self.model = None
        self.graphed = None
        self.fused = None
        self.inflight = {}
        self._gate_event = None
        self._use_registered = False
        self._fifo = []
        self.rr = 0
        self.completed = []
        self._completed_lock = __import__('threading').Lock()
        self.frames = 0
        self.episodes_done = 0
        drain = __import__('threading').Thread(target=self._drain_episodes,
                                               daemon=True)
        drain.start()
        self.timing = {k: 0.0 for k in
                       ('obs', 'fwd', 'sample', 'record', 'env', 'package')}
        self.timing['n'] = 1

    def attach(self, model, device, replay=None):
        """Wire the inference engine (after CUDA init).  In traj mode,
        ``replay`` (a DeviceReplay) receives finished episodes
        device-to-device via commit_traj."""
        self.model = model
        self.device = device
        self.replay = replay
        self.traj = None
        if device.type == 'cuda':
            from .models.geese_net import GeeseFusedEval
            from .hipgraph import GraphedActorForward
            self.fused = GeeseFusedEval(model, device)
            if self.traj_mode:
                assert replay is not None, 'traj mode needs a DeviceReplay'
                from .traj import TrajRecorder
                self.traj = TrajRecorder(
                    self.workers * self.slots * self.n_per, device)
            self.graphed = GraphedActorForward(model, device,
                                               fused=self.fused,
                                               traj=self.traj)
            cap = self.n_per * N_PLAYERS
            mk = lambda shape, dt: [[torch.empty(*shape, dtype=dt,
                                                 pin_memory=True)
                                     for _ in range(self.slots)]
                                    for _ in range(self.workers)]
            self._obs_pin = mk((self.n_per, 17, 7, 11), torch.uint8)
            self._obs_pin_np = [[t.numpy() for t in row] for row in self._obs_pin]
            self._out_pin = mk((cap, 3), torch.float32)
            self._out_pin_np = [[t.numpy() for t in row] for row in self._out_pin]
            self._events = [[torch.cuda.Event() for _ in range(self.slots)]
                            for _ in range(self.workers)]
            if self.traj is not None:
                bucket = self.graphed._bucket(self.n_per)
                self._idx_pin = mk((2, bucket), torch.int64)
                self._idx_pin_np = [[t.numpy() for t in row]
                                    for row in self._idx_pin]
                self._tidx_np = [[np.zeros(self.n_per, dtype=np.int64)
                                  for _ in range(self.slots)]
                                 for _ in range(self.workers)]
            self._register_shm()
            self._svc_graphs = None
            if self.traj is not None:
                self._capture_services()

    def _capture_services(self):
        """Whole-service graphs, one per (worker, slot): H2D (from
        registered shm when available, else the pinned staging buffer) +
        fused forward + sample + in-graph trajectory scatter with
        device-resident step counters + D2H.  A service becomes one
        graph.replay()."""
        import sys
        bucket = self.graphed._bucket(self.n_per)
        dev = self.device
        if os.environ.get('HANDYRL_SVC_GRAPH', '1') != '1':
            return
        graphs, tidxs = [], []
        try:
            for w in range(self.workers):
                grow, trow = [], []
                for s in range(self.slots):
                    base = (w * self.slots + s) * self.n_per
                    gidx = torch.full((bucket,), self.traj.scratch_row,
                                      dtype=torch.int64, device=dev)
                    gidx[:self.n_per] = base + torch.arange(
                        self.n_per, device=dev)
                    tidx = torch.zeros(bucket, dtype=torch.int64, device=dev)
                    # the bundle keeps EVERY tensor the graph reads alive
                    # (graph nodes hold raw pointers; a freed gidx block
                    # reused by a later allocation turns replays into OOB
                    # scatters — the round-2 HSAIL-fault root cause)
                    bundle = self.graphed.capture_service_core(
                        gidx, tidx, self.n_per) + (gidx,)
                    grow.append(bundle)
                    trow.append(tidx)
                graphs.append(grow)
                tidxs.append(trow)
            self._svc_graphs = graphs
            self._tidx_dev = tidxs
            print('# actor service core captured (%dx%d graphs, '
                  'obs source: %s)' % (self.workers, self.slots,
                                       'registered shm'
                                       if self._use_registered
                                       else 'pinned staging'),
                  file=sys.stderr, flush=True)
        except Exception as e:        # noqa: BLE001 - run the op-by-op path
            print('service-pipeline capture failed (%r); per-op path'
                  % (e,), file=sys.stderr, flush=True)
            self._svc_graphs = None

    def _register_shm(self):
        """hipHostRegister the shared-memory obs/result buffers so the DMA
        engines move them directly (no shm->pinned staging memcpy on the
        service path).  Falls back to the staging copy if registration is
        refused (e.g. exotic shm mounts)."""
        self._use_registered = False
        # Round-1 history: direct-DMA from hipHostRegister'd shm was
        # exonerated of the NaN (BASELINE.md root-cause chain) but measured
        # slower on the per-op service path.  With the whole-service graph
        # (traj mode) it removes the staging memcpy from a much smaller
        # service cost, so traj mode defaults it ON; HANDYRL_SHM_REGISTER=0
        # forces staging copies.
        default = '1' if self.traj_mode else '0'
        if os.environ.get('HANDYRL_SHM_REGISTER', default) != '1':
            return
        flat = [v for row in self.obs_views for v in row] + \
               [v for row in self.res_views for v in row]
        try:
            cudart = torch.cuda.cudart()
            registered = []
            for view in flat:
                rc = cudart.cudaHostRegister(view.ctypes.data, view.nbytes, 0)
                if int(rc) != 0:
                    break
                registered.append(view)
            else:
                self._obs_src = [[torch.from_numpy(v) for v in row]
                                 for row in self.obs_views]
                self._res_dst = [[torch.from_numpy(v) for v in row]
                                 for row in self.res_views]
                self._use_registered = all(
                    t.is_pinned()
                    for row in self._obs_src + self._res_dst for t in row)
            if not self._use_registered:
                for view in registered:
                    cudart.cudaHostUnregister(view.ctypes.data)
            else:
                print('# actor shm hipHostRegister: direct-DMA service path',
                      file=sys.stderr)
        except Exception as e:
            print('shm register unavailable (%s); using staged copies' % e,
                  file=sys.stderr)
            self._use_registered = False

    def _drain_episodes(self):
        import multiprocessing.connection as mpc
        while True:
            try:
                ready = mpc.wait(self.ep_conns, timeout=1.0)
            except OSError:
                return
            for conn in ready:
                try:
                    payload = conn.recv()
                except (EOFError, OSError):
                    return
                # workers ship concatenated blocks; split them into
                # zero-copy per-episode views OFF the service path (this
                # background thread), keeping the downstream format
                eps = []
                for item in payload:
                    eps.extend(split_episode_block(item))
                with self._completed_lock:
                    self.completed.extend(eps)
                    self.episodes_done += len(eps)

    def _inflight_cnt(self, wid):
        return sum(1 for (w, _s) in self.inflight if w == wid)

    def _complete(self, wid, slot):
        import time
        t0 = time.time()
        M = self.inflight.pop((wid, slot))
        if M and self.graphed is not None:
            self._events[wid][slot].synchronize()
            if not self._use_registered:
                R = M * N_PLAYERS
                np.copyto(self.res_views[wid][slot][:R],
                          self._out_pin_np[wid][slot][:R])
        self.conns[wid].send(('go', slot))
        self.timing['sample'] += time.time() - t0    # event sync + go

    def _commit_finished(self, base, fin, tidx_dev=None):
        """Traj mode: move finished device-recorded episodes into the
        replay ring (D2D) and queue lightweight stubs for stats.  Runs
        BEFORE issuing the worker's next forward, whose in-graph scatter
        would overwrite these trajectory rows; the main stream waits on
        the commit event to keep that ordering on device.  With
        whole-service graphs, the finished games' device step counters
        are zeroed here (the worker reset them host-side already)."""
        g_local, lens, outcomes = fin
        event = self.replay.commit_traj(self.traj, base + g_local, lens,
                                        outcomes, gate=self._gate_event)
        if event is not None:
            torch.cuda.current_stream().wait_event(event)
        if tidx_dev is not None and len(g_local):
            # blocking H2D: g_local is a temporary pageable array (see
            # replay.commit_traj) — async copy risks garbage indices
            rows = torch.from_numpy(np.ascontiguousarray(g_local)).to(
                self.device)
            tidx_dev.index_fill_(0, rows, 0)
        if self.make_stubs:
            job_args = {'player': list(range(N_PLAYERS)),
                        'model_id': {p: -1 for p in range(N_PLAYERS)}}
            stubs = [{'args': job_args, 'steps': int(lens[k]),
                      'outcome': {p: float(outcomes[k, p])
                                  for p in range(N_PLAYERS)},
                      'committed': True}
                     for k in range(len(g_local))]
            with self._completed_lock:
                self.completed.extend(stubs)
                self.episodes_done += len(stubs)
        else:
            with self._completed_lock:
                self.episodes_done += len(g_local)

    def _poll_completions(self, force_first=False):
        """Complete every in-flight round whose GPU work already finished
        (event fired) — the worker's 'go' goes out the moment its results
        are ready instead of waiting for the next service.  Rounds finish
        in issue order (one engine, one stream), so scanning the FIFO head
        suffices.  ``force_first`` blocks on the oldest round."""
        while self._fifo:
            wid, slot = self._fifo[0]
            M = self.inflight[(wid, slot)]
            if M and self.graphed is not None and not force_first \
                    and not self._events[wid][slot].query():
                break
            self._complete(*self._fifo.pop(0))
            force_first = False

    def step_once(self):
        """Service one child request (whichever is ready first — a
        jittering child never stalls the sweep): collect its obs, issue
        its inference, and hand every finished round back the moment its
        GPU event fires (polled between waits).  With ``slots`` > 1 each
        child keeps another half-shard stepping while this one's round
        trip is in flight.  Returns frames reported."""
        import time
        import multiprocessing.connection as mpc
        t0 = time.time()
        while True:
            self._poll_completions()
            waitable = [c for i, c in enumerate(self.conns)
                        if self._inflight_cnt(i) < self.slots]
            if not waitable:
                self._poll_completions(force_first=True)
                continue
            ready = mpc.wait(waitable, timeout=0.0005 if self._fifo else None)
            if ready:
                break
        conn = self.conns[self.rr] if self.conns[self.rr] in ready else ready[0]
        wid = self.conns.index(conn)
        self.rr = (wid + 1) % self.workers
        msg = conn.recv()
        tag, slot, M, frames = msg[:4]
        assert tag == 'obs'
        self.frames += frames
        self.timing['obs'] += time.time() - t0       # wait + recv
        self.timing['n'] += 1

        t0 = time.time()
        idx_pin = None
        if self.traj is not None:
            _tag, _slot, _m, _f, fin = msg
            base = (wid * self.slots + slot) * self.n_per
            if fin is not None:
                tidx_dev = self._tidx_dev[wid][slot] \
                    if self._svc_graphs is not None else None
                # the worker's previous service event HAS fired (it sent
                # this message only after 'go'): a free, sufficient gate
                self._gate_event = self._events[wid][slot] \
                    if self.graphed is not None else None
                self._commit_finished(base, fin, tidx_dev)
                if self._svc_graphs is None:
                    self._tidx_np[wid][slot][fin[0]] = 0
            if M and self._svc_graphs is not None:
                # service-core graph: forward + sample + trajectory
                # scatter + device step counters as ONE replay; H2D in
                # (from registered shm when available) and D2H out stay
                # eager — ~4 host calls per service
                assert M == self.n_per, (M, self.n_per)
                (graph, static_obs, packed,
                 _zm, _gidx) = self._svc_graphs[wid][slot]
                if self._use_registered:
                    static_obs[:M].copy_(self._obs_src[wid][slot][:M],
                                         non_blocking=True)
                else:
                    np.copyto(self._obs_pin_np[wid][slot][:M],
                              self.obs_views[wid][slot][:M])
                    static_obs[:M].copy_(self._obs_pin[wid][slot][:M],
                                         non_blocking=True)
                graph.replay()
                R = M * N_PLAYERS
                dst = self._res_dst[wid][slot] if self._use_registered \
                    else self._out_pin[wid][slot]
                dst[:R].copy_(packed[:R], non_blocking=True)
                self._events[wid][slot].record()
                self.inflight[(wid, slot)] = M
                self._fifo.append((wid, slot))
                self.timing['fwd'] += time.time() - t0
                if os.environ.get('HANDYRL_ACTOR_SYNC') == '1':
                    while self._fifo:
                        self._complete(*self._fifo.pop(0))
                else:
                    self._poll_completions()
                return frames
            if M:
                # per-op fallback: the parent's mirrored step counters
                # replace the meta the worker used to ship
                assert M == self.n_per, (M, self.n_per)
                bucket = self.graphed._bucket(M)
                idx_np = self._idx_pin_np[wid][slot]
                idx_np[0, :M] = base + np.arange(M)
                idx_np[0, M:bucket] = self.traj.scratch_row
                idx_np[1, :M] = self._tidx_np[wid][slot]
                idx_np[1, M:bucket] = 0
                self._tidx_np[wid][slot] += 1
                idx_pin = self._idx_pin[wid][slot]
        if M and self.graphed is not None:
            if self._use_registered:
                # shm is hipHostRegister'd: DMA straight from/to it, no
                # staging memcpy on the service path
                self.graphed.run_async(self._obs_src[wid][slot], M,
                                       self._res_dst[wid][slot],
                                       self._events[wid][slot],
                                       idx_pinned=idx_pin)
            else:
                np.copyto(self._obs_pin_np[wid][slot][:M],
                          self.obs_views[wid][slot][:M])
                self.graphed.run_async(self._obs_pin[wid][slot], M,
                                       self._out_pin[wid][slot],
                                       self._events[wid][slot],
                                       idx_pinned=idx_pin)
            self.inflight[(wid, slot)] = M
        elif M:
            # CPU fallback (tests): synchronous eager inference on the
            # seat-expanded view (CHMAP gather of the canonical obs)
            canon = self.obs_views[wid][slot][:M].reshape(M, 17, 77)
            obs_np = canon[:, CHMAP].reshape(M * N_PLAYERS, 17, 7, 11)
            obs_t = torch.from_numpy(obs_np.copy()).float()
            with torch.no_grad():
                out = self.model(obs_t, None)
            probs = torch.softmax(out['policy'].float(), dim=-1)
            acts = torch.multinomial(probs, 1).squeeze(-1)
            sel = probs.gather(-1, acts.unsqueeze(-1)).squeeze(-1)
            R = M * N_PLAYERS
            res = self.res_views[wid][slot]
            res[:R, 0] = acts.numpy()
            res[:R, 1] = sel.numpy()
            res[:R, 2] = out['value'].float().squeeze(-1).numpy()
            self.inflight[(wid, slot)] = 0   # results already in shm
        else:
            self.inflight[(wid, slot)] = 0
        self._fifo.append((wid, slot))
        self.timing['fwd'] += time.time() - t0

        # HANDYRL_ACTOR_SYNC=1: complete every service immediately (no
        # overlap at all) — a race-diagnosis probe for
        # tools/repro_transport_race.py
        if os.environ.get('HANDYRL_ACTOR_SYNC') == '1':
            while self._fifo:
                self._complete(*self._fifo.pop(0))
        else:
            self._poll_completions()
        return frames

    def refresh_weights(self):
        if self.fused is not None:
            self.fused.refresh()

    def harvest(self):
        with self._completed_lock:
            out = self.completed
            self.completed = []
        return out

    def shutdown(self):
        while self._fifo:
            self._complete(*self._fifo.pop(0))
        for conn in self.conns:
            try:
                conn.send('quit')
            except (BrokenPipeError, OSError):
                pass
        for proc in self.procs:
            proc.join(timeout=5)
        for shm in self.shms:
            try:
                shm.close()
                shm.unlink()
            except (FileNotFoundError, OSError):
                pass
        self.shms = []
