"""Batched Geister self-play: the recurrent-actor counterpart of
handyrl_amd/actor.py.

Geister is turn-based, partially observed and recurrent (GeisterNet's DRC
Conv-LSTM core), so the reference runs one CPU process per environment
with per-step single-sample CPU inference — the DRC makes that inference
dominate (9 ConvLSTM cell evaluations per step).  Here:

* W child processes each own a shard of python Environment instances and
  do the game-logic side (legal actions, moves, moment recording, episode
  packaging in the reference moment-dict format);
* observations/legal-masks/turn-parity travel through shared memory;
* the parent runs ONE batched GeisterNet forward per child round with the
  per-(game, player) DRC hidden states RESIDENT ON THE GPU — gathered by
  turn parity, scattered back after the step, zeroed on game reset — plus
  the fused masked-softmax-sample kernel over the 214-action space.
"""

import numpy as np
import torch

from . import ops
from .environment import make_env
from .hipgraph import CAPTURE_LOCK

N_ACTIONS = 214
SCALAR_DIM = 18
BOARD_SHAPE = (7, 6, 6)
REC_CAP = 202                            # 2 layout turns + 200 moves
_JOB_ARGS = {'player': [0, 1], 'model_id': {0: -1, 1: -1}}


def _make_rec_buffers(G):
    """Columnar per-game recording buffers (turn-based episode format)."""
    rec = {
        'scalar': np.zeros((G, REC_CAP, SCALAR_DIM), np.uint8),
        'board': np.zeros((G, REC_CAP) + BOARD_SHAPE, np.uint8),
        'mask': np.zeros((G, REC_CAP, N_ACTIONS), bool),
        'turn': np.zeros((G, REC_CAP), np.int8),
        'action': np.zeros((G, REC_CAP), np.int16),
        'prob': np.zeros((G, REC_CAP), np.float32),
        'value': np.zeros((G, REC_CAP), np.float32),
    }
    return rec, np.zeros(G, np.int32)


def _record_round(rec, rec_len, gar, scalar, board, mask, parity, actions,
                  probs, values):
    """Scatter one step of every game into the columnar buffers."""
    rows = np.minimum(rec_len, REC_CAP - 1)
    rec['scalar'][gar, rows] = scalar.astype(np.uint8)
    rec['board'][gar, rows] = board.astype(np.uint8)
    rec['mask'][gar, rows] = mask == 0.0
    rec['turn'][gar, rows] = parity
    rec['action'][gar, rows] = actions.astype(np.int16)
    rec['prob'][gar, rows] = probs
    rec['value'][gar, rows] = values
    rec_len += 1


def _package_columnar(rec, g, S, outcome, gamma):
    """One game's recording -> a columnar turn-based episode dict."""
    if S == 0:
        return None
    acc, rets = 0.0, np.empty(S, np.float32)
    for t in range(S - 1, -1, -1):       # constant -0.01 both players
        acc = -0.01 + gamma * acc
        rets[t] = acc
    ep = {'args': _JOB_ARGS, 'steps': S, 'outcome': outcome,
          'columnar': True, 'turn_based': True,
          'n_actions': N_ACTIONS, 'n_players': 2,
          'reward': np.full((S, 2), -0.01, np.float32),
          'return': np.stack([rets, rets], axis=1)}
    for k, buf in rec.items():
        ep[k] = buf[g, :S].copy()
    return ep


def _geister_env_worker(conn, ep_conn, shm_names, n_games, args, seed):
    from multiprocessing import shared_memory
    from .batch import pack_moments

    shms = {k: shared_memory.SharedMemory(name=v) for k, v in shm_names.items()}
    G = n_games
    scalar_v = np.ndarray((G, SCALAR_DIM), dtype=np.float32, buffer=shms['scalar'].buf)
    board_v = np.ndarray((G,) + BOARD_SHAPE, dtype=np.float32, buffer=shms['board'].buf)
    mask_v = np.ndarray((G, N_ACTIONS), dtype=np.float32, buffer=shms['mask'].buf)
    parity_v = np.ndarray((G,), dtype=np.int8, buffer=shms['parity'].buf)
    reset_v = np.ndarray((G,), dtype=np.uint8, buffer=shms['reset'].buf)
    res_v = np.ndarray((G, 4), dtype=np.float32, buffer=shms['res'].buf)

    import random
    random.seed(seed)
    envs = [make_env({'env': 'Geister', 'id': seed * 1000 + i}) for i in range(G)]
    for env in envs:
        env.reset()
    moments = [[] for _ in range(G)]
    gamma = args.get('gamma', 0.8)
    compress = args.get('compress_episodes', False)
    compress_steps = args.get('compress_steps', 4)
    job_args = {'player': [0, 1], 'model_id': {0: -1, 1: -1}}

    def package(g):
        ms = moments[g]
        if not ms:
            return None
        outcome = envs[g].outcome()
        for p in (0, 1):
            ret = 0.0
            for m in reversed(ms):
                ret = (m['reward'][p] or 0) + gamma * ret
                m['return'][p] = ret
        return {'args': job_args, 'steps': len(ms), 'outcome': outcome,
                'moment': pack_moments(ms, compress_steps, compress=compress)}

    frames_prev = 0
    eps_out = []
    while True:
        # stage one observation round
        for g, env in enumerate(envs):
            reset_v[g] = 0
            if env.terminal():
                ep = package(g)
                if ep is not None:
                    eps_out.append(ep)
                moments[g] = []
                env.reset()
                reset_v[g] = 1
            p = env.turn()
            obs = env.observation(p)
            scalar_v[g] = obs['scalar']
            board_v[g] = obs['board']
            legal = env.legal_actions(p)
            mask_v[g] = 1e32
            mask_v[g, legal] = 0.0
            parity_v[g] = p

        if eps_out:
            # episodes travel on their own pipe, drained by a parent
            # background thread: the service path never deserializes them
            ep_conn.send(eps_out)
        conn.send(('obs', G, frames_prev))
        frames_prev, eps_out = 0, []
        cmd = conn.recv()
        if cmd == 'quit':
            break

        # apply sampled actions + record moments (reference generation.py
        # semantics for the turn-based, observation=False configuration)
        for g, env in enumerate(envs):
            p = int(parity_v[g])
            action = int(res_v[g, 0])
            moment = {key: {0: None, 1: None} for key in
                      ('observation', 'selected_prob', 'action_mask', 'action',
                       'value', 'reward', 'return')}
            moment['observation'][p] = {'scalar': scalar_v[g].copy(),
                                        'board': board_v[g].copy()}
            moment['selected_prob'][p] = float(res_v[g, 1])
            moment['action_mask'][p] = mask_v[g].copy()
            moment['action'][p] = action
            moment['value'][p] = np.array([res_v[g, 2]], dtype=np.float32)
            moment['turn'] = [p]
            env.play(action)
            reward = env.reward()
            for q in (0, 1):
                moment['reward'][q] = reward.get(q, None)
            moments[g].append(moment)
            frames_prev += 1


def _geister_vec_worker(conn, ep_conn, shm_names, n_games, args, seed,
                        traj_mode=False):
    """Vectorized variant of _geister_env_worker: one GeisterVecEnv steps
    the whole shard (legality, observations, moves, captures and win
    detection as batched numpy ops — envs/vec_geister.py, parity-tested
    against the single-game oracle); only the per-step moment recording
    stays per-game python, in the reference episode format."""
    from multiprocessing import shared_memory
    from .batch import pack_moments
    from .envs.vec_geister import GeisterVecEnv

    shms = {k: shared_memory.SharedMemory(name=v) for k, v in shm_names.items()}
    G = n_games
    scalar_v = np.ndarray((G, SCALAR_DIM), dtype=np.float32, buffer=shms['scalar'].buf)
    board_v = np.ndarray((G,) + BOARD_SHAPE, dtype=np.float32, buffer=shms['board'].buf)
    mask_v = np.ndarray((G, N_ACTIONS), dtype=np.float32, buffer=shms['mask'].buf)
    parity_v = np.ndarray((G,), dtype=np.int8, buffer=shms['parity'].buf)
    reset_v = np.ndarray((G,), dtype=np.uint8, buffer=shms['reset'].buf)
    res_v = np.ndarray((G, 4), dtype=np.float32, buffer=shms['res'].buf)

    import os
    vec = GeisterVecEnv(G, seed=seed)
    vec.reset_games(np.arange(G))
    gamma = args.get('gamma', 0.8)
    compress = args.get('compress_episodes', False)
    compress_steps = args.get('compress_steps', 4)
    job_args = {'player': [0, 1], 'model_id': {0: -1, 1: -1}}
    reward = {0: -0.01, 1: -0.01}            # geister.py reward(): constant
    # columnar episodes (plain arrays consumed by make_batch's turn-based
    # columnar fast path; parity-tested against the moment-dict format in
    # tests/test_columnar_turn.py) skip all per-step dict building —
    # HANDYRL_GEISTER_COLUMNAR=0 restores reference-format moments
    columnar = os.environ.get('HANDYRL_GEISTER_COLUMNAR', '1') == '1'
    moments = [[] for _ in range(G)]
    gar = np.arange(G)
    if traj_mode:
        rec, rec_len = None, np.zeros(G, np.int32)   # counters only
    elif columnar:
        rec, rec_len = _make_rec_buffers(G)

    def package(g, outcome):
        if columnar:
            return _package_columnar(rec, g, int(rec_len[g]), outcome, gamma)
        ms = moments[g]
        if not ms:
            return None
        for p in (0, 1):
            ret = 0.0
            for m in reversed(ms):
                ret = (m['reward'][p] or 0) + gamma * ret
                m['return'][p] = ret
        return {'args': job_args, 'steps': len(ms), 'outcome': outcome,
                'moment': pack_moments(ms, compress_steps, compress=compress)}

    frames_prev = 0
    eps_out = []
    fin = None
    while True:
        reset_v[:] = 0
        done_idx = np.nonzero(vec.over)[0]
        if len(done_idx):
            ocs = vec.outcomes(done_idx)
            if traj_mode:
                # fin metadata only: the parent commits the device-recorded
                # trajectories; lens exclude rows past the recorder cap
                lens = np.minimum(rec_len[done_idx], REC_CAP).astype(np.int64)
                fin = (done_idx.astype(np.int64), lens,
                       ocs.astype(np.float32))
                rec_len[done_idx] = 0
            else:
                for k, g in enumerate(done_idx):
                    ep = package(g, {0: float(ocs[k, 0]),
                                     1: float(ocs[k, 1])})
                    if ep is not None:
                        eps_out.append(ep)
                    if columnar:
                        rec_len[g] = 0
                    else:
                        moments[g] = []
            vec.reset_games(done_idx)
            reset_v[done_idx] = 1
        scalar, board = vec.observations()
        scalar_v[:] = scalar
        board_v[:] = board
        vec.legal_masks(out=mask_v)
        parity_v[:] = vec.turn()

        if eps_out:
            # episodes travel on their own pipe, drained by a parent
            # background thread: the service path never deserializes them
            ep_conn.send(eps_out)
        if traj_mode:
            conn.send(('obs', G, frames_prev, fin))
            fin = None
        else:
            conn.send(('obs', G, frames_prev))
        frames_prev, eps_out = 0, []
        cmd = conn.recv()
        if cmd == 'quit':
            break

        actions = res_v[:, 0].astype(np.int64)
        # visibility: sampled actions that violate the mask we shipped
        # (a NaN policy row or an all-illegal row samples arbitrarily;
        # the env treats them as a pass, but they signal upstream NaN)
        bad = (actions < 0) | (actions >= N_ACTIONS)
        ok = ~bad
        bad[ok] |= mask_v[gar[ok], actions[ok]] != 0.0
        nbad = int(bad.sum())
        if nbad:
            import sys as _sys
            print('# geister worker: %d illegal sampled actions'
                  % nbad, file=_sys.stderr, flush=True)
        if traj_mode:
            rec_len += 1               # step counter only (device records)
        elif columnar:
            _record_round(rec, rec_len, gar, scalar_v, board_v, mask_v,
                          parity_v, actions, res_v[:, 1], res_v[:, 2])
        else:
            for g in range(G):
                p = int(parity_v[g])
                moment = {key: {0: None, 1: None} for key in
                          ('observation', 'selected_prob', 'action_mask',
                           'action', 'value', 'reward', 'return')}
                moment['observation'][p] = {'scalar': scalar_v[g].copy(),
                                            'board': board_v[g].copy()}
                moment['selected_prob'][p] = float(res_v[g, 1])
                moment['action_mask'][p] = mask_v[g].copy()
                moment['action'][p] = int(actions[g])
                moment['value'][p] = np.array([res_v[g, 2]], dtype=np.float32)
                moment['turn'] = [p]
                moment['reward'] = dict(reward)
                moments[g].append(moment)
        vec.step(actions)
        frames_prev += G


class BatchedDRCEngine:
    """Batched recurrent (DRC Conv-LSTM) inference for one shard of games:
    per-(game, player) hidden state RESIDENT on the device, gathered by
    turn parity per step, scattered back after, zeroed on game reset.
    hipGraph-captured on GPU (in-place hidden updates keep the storages
    fixed under capture); eager on CPU / when capture fails."""

    def __init__(self, model, device, n_games, use_graphs=True,
                 traj=None, traj_base=0):
        import os
        self.model = model
        self.device = device
        self.n_per = n_games
        # device-side trajectory recording (handyrl_amd/traj
        # GeisterTrajRecorder): the captured service scatters the mover's
        # scalar/board/mask/turn and the packed outputs into HBM rings
        self.traj = traj
        if traj is not None:
            self._gidx = traj_base + torch.arange(n_games, dtype=torch.int64,
                                                  device=device)
            self.tidx = torch.zeros(n_games, dtype=torch.int64, device=device)
        # fused DRC: one hand-written CDNA4 kernel per ConvLSTM cell eval
        # (ops/src/ext.hip::convlstm_cell) with NHWC bf16 hidden state
        # (HANDYRL_DRC_FUSED=0 opts back into the eager cell chain)
        self.fused_drc = (device.type == 'cuda' and ops.available() and
                          os.environ.get('HANDYRL_DRC_FUSED', '1') == '1')
        if self.fused_drc:
            G2 = n_games * 2
            self.hidden = (
                [torch.zeros(G2, 36, 32, dtype=torch.bfloat16, device=device)
                 for _ in range(3)],
                [torch.zeros(G2, 36, 32, dtype=torch.float32, device=device)
                 for _ in range(3)])
            self._nbr = ops.convlstm_neighbor_table(device)
            self._wfrag = [torch.zeros(2, 9, 8, 4, 16, 8,
                                       dtype=torch.bfloat16, device=device)
                           for _ in range(3)]
            self._wbias = [torch.zeros(128, device=device) for _ in range(3)]
            self.refresh()
        else:
            hs, cs = model.init_hidden([n_games * 2])
            self.hidden = ([h.to(device) for h in hs],
                           [c.to(device) for c in cs])
        self._arange2 = torch.arange(n_games, device=device) * 2
        self._graph = None
        if use_graphs and device.type == 'cuda' and \
                __import__('os').environ.get('HANDYRL_NO_GRAPHS') != '1':
            self._capture()

    @torch.no_grad()
    def refresh(self):
        """Repack the cell conv weights into MFMA fragments (call after a
        model-weight push; the packed buffers are graph-stable)."""
        if not self.fused_drc:
            return
        for l, cell in enumerate(self.model.body.blocks):
            self._wfrag[l].copy_(
                ops.pack_convlstm_weights(cell.conv.weight.detach()))
            self._wbias[l].copy_(cell.conv.bias.detach().float())

    def _static_in(self):
        per, dev = self.n_per, self.device
        st = {
            'scalar': torch.zeros(per, SCALAR_DIM, device=dev),
            'board': torch.zeros((per,) + BOARD_SHAPE, device=dev),
            'mask': torch.full((per, N_ACTIONS), 1e32, device=dev),
            'parity': torch.zeros(per, dtype=torch.int64, device=dev),
            'keep': torch.ones(per * 2, 1, 1, 1, device=dev),
        }
        if self.fused_drc:
            st['keep'] = torch.ones(per * 2, 1, 1, device=dev)
            st['keep_h'] = torch.ones(per * 2, 1, 1, dtype=torch.bfloat16,
                                      device=dev)
        return st

    def _drc_fused(self, x_nchw, rows):
        """The DRC core on the fused cell kernel: 3 layers x 3 repeats,
        ping-pong buffers, per-(game, player) hidden gathered by parity."""
        hs, cs = self.hidden
        B = x_nchw.shape[0]
        x = x_nchw.permute(0, 2, 3, 1).reshape(B, 36, 32) \
            .to(torch.bfloat16).contiguous()
        h_cur = [h.index_select(0, rows).contiguous() for h in hs]
        c_cur = [c.index_select(0, rows).contiguous() for c in cs]
        h_alt = [torch.empty_like(t) for t in h_cur]
        c_alt = [torch.empty_like(t) for t in c_cur]
        for _rep in range(3):
            for l in range(3):
                src = x if l == 0 else h_alt[l - 1]
                ops.convlstm_cell(src, h_cur[l], c_cur[l], self._wfrag[l],
                                  self._wbias[l], self._nbr,
                                  h_alt[l], c_alt[l])
            h_cur, h_alt = h_alt, h_cur
            c_cur, c_alt = c_alt, c_cur
        for i in range(3):
            hs[i].index_copy_(0, rows, h_cur[i])
            cs[i].index_copy_(0, rows, c_cur[i])
        return h_cur[2].float().reshape(B, 6, 6, 32) \
            .permute(0, 3, 1, 2).contiguous()

    def _infer_body(self, st):
        """One batched recurrent step on static tensors; hidden updates are
        IN-PLACE so the storages stay fixed under graph capture."""
        import torch.nn.functional as F
        from .models.common import apply_bn
        hs, cs = self.hidden
        rows = self._arange2 + st['parity']
        if self.fused_drc:
            model = self.model
            for i in range(len(hs)):
                hs[i].mul_(st['keep_h'])
                cs[i].mul_(st['keep'])
            B = st['scalar'].shape[0]
            planes = st['scalar'].view(B, SCALAR_DIM, 1, 1) \
                .expand(B, SCALAR_DIM, 6, 6)
            stem = F.relu(apply_bn(model.bn1, model.conv1(
                torch.cat([planes, st['board']], dim=1))))
            h_last = self._drc_fused(stem, rows)
            p_move = model.head_p_move(h_last)
            p_set = model.head_p_set(st['scalar'][:, :1])
            policy = torch.cat([p_move, p_set], dim=-1).float()
            value = torch.tanh(model.head_v(h_last)).float()
            ret = model.head_r(h_last).float()
        else:
            for i in range(len(hs)):
                hs[i].mul_(st['keep'])
                cs[i].mul_(st['keep'])
            h_in = ([h.index_select(0, rows) for h in hs],
                    [c.index_select(0, rows) for c in cs])
            out = self.model({'scalar': st['scalar'],
                              'board': st['board']}, h_in)
            h_out, c_out = out['hidden']
            for i in range(len(hs)):
                hs[i].index_copy_(0, rows, h_out[i])
                cs[i].index_copy_(0, rows, c_out[i])
            policy = out['policy'].float()
            value = out['value'].float()
            ret = out['return'].float()
        uniform = torch.rand(policy.shape[0], device=self.device)
        if self.device.type == 'cuda':
            actions, probs = ops.masked_sample(policy, st['mask'], uniform)
        else:
            pr = torch.softmax(policy - st['mask'], dim=-1)
            actions = torch.multinomial(pr, 1).squeeze(-1)
            probs = pr.gather(-1, actions.unsqueeze(-1)).squeeze(-1)
        packed = torch.stack([actions.float(), probs,
                              value.squeeze(-1),
                              ret.squeeze(-1)], dim=1)
        if self.traj is not None:
            self.traj.record_(st['scalar'], st['board'], st['mask'],
                              st['parity'], packed, self._gidx, self.tidx)
            self.tidx.add_(1).clamp_(0, self.traj.max_steps - 1)
        return packed

    @torch.no_grad()
    def _capture(self):
        try:
            st = self._static_in()
            stream = torch.cuda.Stream()
            stream.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(stream):
                for _ in range(2):
                    self._infer_body(st)
            torch.cuda.current_stream().wait_stream(stream)
            graph = torch.cuda.CUDAGraph()
            with CAPTURE_LOCK, torch.cuda.graph(
                graph, capture_error_mode='thread_local'):
                packed = self._infer_body(st)
            self._graph = (graph, st, packed)
            # captured warmups corrupted the hidden state: reset it
            hs, cs = self.hidden
            for t in hs + cs:
                t.zero_()
            if self.traj is not None:
                self.tidx.zero_()     # warmups advanced the counters
        except Exception as e:     # noqa: BLE001 - eager fallback
            import sys
            print('geister actor graph capture failed, running eager: %r'
                  % (e,), file=sys.stderr)
            self._graph = None

    @torch.no_grad()
    def infer_async(self, scalar, board, mask, parity, reset, out_pin,
                    event):
        """Issue one batched recurrent step WITHOUT a host sync: the
        packed (action, prob, value, return) rows land in pinned
        ``out_pin`` and ``event`` records completion.  Returns False when
        no captured graph is available (caller falls back to infer()).
        scalar/board/mask must outlive the copy (they are shm views the
        worker will not touch until 'go'); the parity/keep temporaries
        are copied blocking."""
        if self._graph is None:
            return False
        per = self.n_per
        shape = (per * 2, 1, 1) if self.fused_drc else (per * 2, 1, 1, 1)
        keep_np = (1.0 - reset.astype(np.float32)).repeat(2).reshape(shape)
        graph, st, packed = self._graph
        st['scalar'].copy_(torch.from_numpy(scalar), non_blocking=True)
        st['board'].copy_(torch.from_numpy(board), non_blocking=True)
        st['mask'].copy_(torch.from_numpy(mask), non_blocking=True)
        st['parity'].copy_(torch.from_numpy(parity.astype(np.int64)))
        st['keep'].copy_(torch.from_numpy(keep_np))
        if self.fused_drc:
            st['keep_h'].copy_(st['keep'])
        graph.replay()
        out_pin.copy_(packed, non_blocking=True)
        event.record()
        return True

    @torch.no_grad()
    def infer(self, scalar, board, mask, parity, reset, out):
        """numpy in (shard-sized arrays), numpy out: fills out[:, 0:4] with
        (action, prob, value, return) rows."""
        per, dev = self.n_per, self.device
        shape = (per * 2, 1, 1) if self.fused_drc else (per * 2, 1, 1, 1)
        keep_np = (1.0 - reset.astype(np.float32)).repeat(2).reshape(shape)
        if self._graph is not None:
            graph, st, packed = self._graph
            # scalar/board/mask are shm views (stable until 'go'), but
            # parity.astype and keep_np are TEMPORARIES: all copies stay
            # blocking — hipMemcpyAsync from pageable memory can outlive
            # a freed source and read garbage
            st['scalar'].copy_(torch.from_numpy(scalar))
            st['board'].copy_(torch.from_numpy(board))
            st['mask'].copy_(torch.from_numpy(mask))
            st['parity'].copy_(torch.from_numpy(parity.astype(np.int64)))
            st['keep'].copy_(torch.from_numpy(keep_np))
            if self.fused_drc:
                st['keep_h'].copy_(st['keep'])
            graph.replay()
            np.copyto(out, packed.cpu().numpy())
            return
        st = {
            'scalar': torch.from_numpy(scalar).to(dev),
            'board': torch.from_numpy(board).to(dev),
            'mask': torch.from_numpy(mask).to(dev),
            'parity': torch.from_numpy(parity.astype(np.int64)).to(dev),
            'keep': torch.from_numpy(keep_np).to(dev),
        }
        if self.fused_drc:
            st['keep_h'] = st['keep'].to(torch.bfloat16)
        np.copyto(out, self._infer_body(st).cpu().numpy())


class GeisterActorPool:
    """In-process Geister self-play (the `worker: {type: 'gpu'}` pool for
    recurrent envs): the vectorized rules engine steps the whole shard in
    the calling thread, batched DRC inference runs with device-resident
    hidden state (BatchedDRCEngine), and episodes are recorded columnar
    turn-based.  No worker processes — safe to start from a thread of an
    already-CUDA-initialized learner (unlike a forking pool)."""

    def __init__(self, model, args, n_games=256, device=None, seed=0):
        self.args = args
        self.device = device if device is not None else (
            torch.device('cuda') if torch.cuda.is_available()
            else torch.device('cpu'))
        self.model = model
        from .envs.vec_geister import GeisterVecEnv
        self.vec = GeisterVecEnv(n_games, seed=seed)
        self.vec.reset_games(np.arange(n_games))
        G = n_games
        self.n_games = G
        self.gamma = args.get('gamma', 0.8)
        self.rec, self.rec_len = _make_rec_buffers(G)
        self._gar = np.arange(G)
        self._mask_v = np.empty((G, N_ACTIONS), np.float32)
        self._res_v = np.empty((G, 4), np.float32)
        self.engine = BatchedDRCEngine(model, self.device, G)
        self.completed = []
        self.frames = 0
        self.episodes_done = 0
        self.calls_per_vec_step = 1

    @torch.no_grad()
    def step_once(self):
        vec = self.vec
        G = self.n_games
        reset_flags = np.zeros(G, np.uint8)
        done_idx = np.nonzero(vec.over)[0]
        if len(done_idx):
            ocs = vec.outcomes(done_idx)
            for k, g in enumerate(done_idx):
                ep = _package_columnar(self.rec, g, int(self.rec_len[g]),
                                       {0: float(ocs[k, 0]),
                                        1: float(ocs[k, 1])}, self.gamma)
                if ep is not None:
                    self.completed.append(ep)
                    self.episodes_done += 1
                self.rec_len[g] = 0
            vec.reset_games(done_idx)
            reset_flags[done_idx] = 1
        scalar, board = vec.observations()
        vec.legal_masks(out=self._mask_v)
        parity = vec.turn().astype(np.int8)
        self.engine.infer(scalar, board, self._mask_v, parity, reset_flags,
                          self._res_v)
        actions = self._res_v[:, 0].astype(np.int64)
        _record_round(self.rec, self.rec_len, self._gar, scalar, board,
                      self._mask_v, parity, actions, self._res_v[:, 1],
                      self._res_v[:, 2])
        vec.step(actions)
        self.frames += G
        return G

    def harvest(self):
        out = self.completed
        self.completed = []
        return out

    def refresh_weights(self):
        self.engine.refresh()      # repack fused-DRC weight fragments

    def shutdown(self):
        pass


class GeisterMultiProcPool:
    """256-actor-style Geister self-play on one GPU: W env-worker processes
    + batched recurrent inference with GPU-resident DRC hidden state."""

    def __init__(self, args, n_games=256, seed=0, workers=8, vec=None,
                 traj_mode=False, make_stubs=True):
        import multiprocessing as mp
        import os
        from multiprocessing import shared_memory
        self.args = args
        self.workers = workers
        # traj_mode: GeisterTrajRecorder rings + TurnDeviceReplay.commit_traj
        # (requires attach(..., replay=TurnDeviceReplay) on CUDA): workers
        # ship only fin metadata, never episode arrays
        self.traj_mode = traj_mode
        self.make_stubs = make_stubs
        if vec is None:
            vec = os.environ.get('HANDYRL_GEISTER_VEC', '1') == '1'
        if traj_mode:
            vec = True                  # traj workers are the vec variant
        worker_fn = _geister_vec_worker if vec else _geister_env_worker
        per = max(1, n_games // workers)
        self.n_per = per
        self.conns, self.ep_conns, self.procs, self.shms = [], [], [], []
        self.views = []
        sizes = {
            'scalar': per * SCALAR_DIM * 4,
            'board': per * int(np.prod(BOARD_SHAPE)) * 4,
            'mask': per * N_ACTIONS * 4,
            'parity': per,
            'reset': per,
            'res': per * 4 * 4,
        }
        for w in range(workers):
            shm = {k: shared_memory.SharedMemory(create=True, size=v)
                   for k, v in sizes.items()}
            self.shms.extend(shm.values())
            views = {
                'scalar': np.ndarray((per, SCALAR_DIM), dtype=np.float32,
                                     buffer=shm['scalar'].buf),
                'board': np.ndarray((per,) + BOARD_SHAPE, dtype=np.float32,
                                    buffer=shm['board'].buf),
                'mask': np.ndarray((per, N_ACTIONS), dtype=np.float32,
                                   buffer=shm['mask'].buf),
                'parity': np.ndarray((per,), dtype=np.int8,
                                     buffer=shm['parity'].buf),
                'reset': np.ndarray((per,), dtype=np.uint8,
                                    buffer=shm['reset'].buf),
                'res': np.ndarray((per, 4), dtype=np.float32,
                                  buffer=shm['res'].buf),
            }
            self.views.append(views)
            parent_conn, child_conn = mp.Pipe(duplex=True)
            ep_parent, ep_child = mp.Pipe(duplex=False)
            wargs = (child_conn, ep_child,
                     {k: s.name for k, s in shm.items()},
                     per, args, seed + 131 * w)
            if worker_fn is _geister_vec_worker:
                wargs = wargs + (traj_mode,)
            proc = mp.Process(target=worker_fn, args=wargs, daemon=True)
            proc.start()
            child_conn.close()
            ep_child.close()
            self.conns.append(parent_conn)
            self.ep_conns.append(ep_parent)
            self.procs.append(proc)

        self.model = None
        self.device = None
        self.engines = None
        self.completed = []
        self._completed_lock = __import__('threading').Lock()
        self.frames = 0
        self.episodes_done = 0
        self.rr = 0
        self.calls_per_vec_step = workers
        drain = __import__('threading').Thread(target=self._drain_episodes,
                                               daemon=True)
        drain.start()

    def attach(self, model, device, replay=None):
        import os
        self.model = model
        self.device = device
        self.replay = replay
        self.traj = None
        if self.traj_mode:
            assert replay is not None, 'traj mode needs a TurnDeviceReplay'
            from .traj import GeisterTrajRecorder
            self.traj = GeisterTrajRecorder(self.workers * self.n_per,
                                            device)
        # one engine (resident hidden + captured graph) per worker shard
        self.engines = [BatchedDRCEngine(model, device, self.n_per,
                                         traj=self.traj,
                                         traj_base=w * self.n_per)
                        for w in range(self.workers)]
        # event-polled async service (the geese-pool design): the DRC
        # round trip of one worker overlaps every other worker's env work;
        # HANDYRL_GEISTER_ASYNC=0 restores the synchronous service
        self._async = (device.type == 'cuda'
                       and os.environ.get('HANDYRL_GEISTER_ASYNC', '1') == '1')
        if self._async:
            self._out_pin = [torch.empty(self.n_per, 4, dtype=torch.float32,
                                         pin_memory=True)
                             for _ in range(self.workers)]
            self._out_pin_np = [t.numpy() for t in self._out_pin]
            self._events = [torch.cuda.Event() for _ in range(self.workers)]
        self._fifo = []

    @torch.no_grad()
    def _infer(self, wid):
        v = self.views[wid]
        self.engines[wid].infer(v['scalar'], v['board'], v['mask'],
                                v['parity'], v['reset'], v['res'])

    def _commit_finished(self, wid, fin):
        """Traj mode: commit finished device-recorded episodes into the
        replay ring and zero the engine's step counters for those games
        (ordering: the engine's next replay waits on the commit event)."""
        g_local, lens, outcomes = fin
        base = wid * self.n_per
        gate = self._events[wid] if self._async else None
        event = self.replay.commit_traj(self.traj, base + g_local, lens,
                                        outcomes, gate=gate)
        if event is not None:
            torch.cuda.current_stream().wait_event(event)
        if len(g_local):
            rows = torch.from_numpy(np.ascontiguousarray(g_local)).to(
                self.device)          # blocking: temporary pageable source
            self.engines[wid].tidx.index_fill_(0, rows, 0)
        if self.make_stubs:
            stubs = [{'args': _JOB_ARGS, 'steps': int(lens[k]),
                      'outcome': {0: float(outcomes[k, 0]),
                                  1: float(outcomes[k, 1])},
                      'committed': True}
                     for k in range(len(g_local))]
            with self._completed_lock:
                self.completed.extend(stubs)
                self.episodes_done += len(stubs)
        else:
            with self._completed_lock:
                self.episodes_done += len(g_local)

    def _complete(self, wid):
        self._events[wid].synchronize()
        np.copyto(self.views[wid]['res'], self._out_pin_np[wid])
        self.conns[wid].send('go')

    def _poll_completions(self, force_first=False):
        """'go' goes out the moment a worker's DRC results are ready
        (events fire in issue order: one model, one stream)."""
        while self._fifo:
            wid = self._fifo[0]
            if not force_first and not self._events[wid].query():
                break
            self._complete(self._fifo.pop(0))
            force_first = False

    def step_once(self):
        import multiprocessing.connection as mpc
        if not self._async:
            ready = mpc.wait(self.conns)
            conn = self.conns[self.rr] \
                if self.conns[self.rr] in ready else ready[0]
            wid = self.conns.index(conn)
            self.rr = (wid + 1) % self.workers
            msg = conn.recv()
            tag, G, frames = msg[:3]
            assert tag == 'obs'
            self.frames += frames
            if self.traj is not None and msg[3] is not None:
                self._commit_finished(wid, msg[3])
            self._infer(wid)
            conn.send('go')
            return frames

        inflight = set(self._fifo)
        while True:
            self._poll_completions()
            inflight = set(self._fifo)
            waitable = [c for i, c in enumerate(self.conns)
                        if i not in inflight]
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
        tag, G, frames = msg[:3]
        assert tag == 'obs'
        self.frames += frames
        if self.traj is not None and msg[3] is not None:
            self._commit_finished(wid, msg[3])
        v = self.views[wid]
        if self.engines[wid].infer_async(v['scalar'], v['board'], v['mask'],
                                         v['parity'], v['reset'],
                                         self._out_pin[wid],
                                         self._events[wid]):
            self._fifo.append(wid)
            self._poll_completions()
        else:                      # capture unavailable: synchronous path
            self._infer(wid)
            conn.send('go')
        return frames

    def _drain_episodes(self):
        import multiprocessing.connection as mpc
        while True:
            try:
                ready = mpc.wait(self.ep_conns, timeout=1.0)
            except OSError:
                return
            for conn in ready:
                try:
                    eps = conn.recv()
                except (EOFError, OSError):
                    return
                with self._completed_lock:
                    self.completed.extend(eps)
                    self.episodes_done += len(eps)

    def harvest(self):
        with self._completed_lock:
            out = self.completed
            self.completed = []
        return out

    def refresh_weights(self):
        if self.engines:
            for e in self.engines:
                e.refresh()        # repack fused-DRC weight fragments

    def shutdown(self):
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
