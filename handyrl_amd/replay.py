"""Device-resident trajectory replay for columnar episodes.

The MI355X replacement for the host-side episode deque + multiprocess
batch builders on the flagship (solo, feed-forward) path: trajectories
live in a flat columnar ring in HBM3E (capacity is a byte budget — 288 GB
per GPU makes very large replays practical), window sampling keeps the
reference's recency-biased accept-loop semantics on the host (tiny index
math), and the (B, T, 1, ...) batch is GATHERED ON DEVICE with ~20 torch
ops that capture into the training hipGraph — so one graph replay does
sample-gather + forward + loss + backward + all-reduce + Adam with no
per-step host batch traffic.

Two ring flavors share the machinery (ColumnRingReplay): DeviceReplay
(feed-forward solo, the flagship Hungry Geese path) and TurnDeviceReplay
(turn-based/recurrent, Geister, incl. burn-in windows).  Semantics parity
with batch.make_batch is tested in tests/test_replay.py and
tests/test_turn_replay.py (CPU) and on GPU in tests/test_gpu.py.
"""

import queue as queue_mod
import random
import threading
from collections import deque

import numpy as np
import torch

# ring rows store CANONICAL per-game obs (one 17-plane board per step, not
# four seat views): 4x the episode capacity per byte of HBM; the per-seat
# channel rotation (vec_geese.CHMAP) happens inside gather_batch on device
OBS_SHAPE = (17, 7, 11)
ROW_OBS = int(np.prod(OBS_SHAPE))

# sampleable rows keep this distance from the overwrite frontier so that
# in-flight ring writes (background ingest stream) can never race a graph
# replay's gathers of still-tabled episodes
EVICT_MARGIN = 16384


class ColumnRingReplay:
    """Flat columnar ring of per-step rows + a host-side episode table.

    Subclasses declare ``COLUMNS`` ({attr: (torch dtype, tail shape,
    episode key)}) and ``OUTCOME_P`` and provide env-shape-specific
    ``sample_indices`` / ``gather_batch``; everything else — staging,
    ring writes, the background-ingest thread + event-ordered publish,
    planned-frontier eviction and reader-floor back-pressure — is shared.

    With ``ingest_thread=True`` (the bench default on GPU) episode blocks
    are staged and copied on a background thread + side HIP stream; the
    consumer calls ``publish()`` (done inside GraphedReplayTrainStep._fill)
    to make completed writes sampleable, with a stream wait on the write
    event so gathers are ordered after the copies.
    """

    COLUMNS = {}
    OUTCOME_P = 2

    def __init__(self, args, device, bytes_budget=4 << 30, ingest_thread=False):
        self.args = args
        self.device = device
        row_bytes = sum(
            int(np.prod(tail or (1,))) * torch.empty((), dtype=dt).element_size()
            for dt, tail, _k in self.COLUMNS.values())
        self.ring_T = max(1024, int(bytes_budget // row_bytes))
        for name, (dt, tail, _k) in self.COLUMNS.items():
            setattr(self, name, torch.empty((self.ring_T,) + tail, dtype=dt,
                                            device=device))
        self.head = 0                  # monotonically increasing write cursor
        self.head_planned = 0          # head + queued-but-unwritten rows
        # oldest logical row any in-flight gather may read; None until the
        # first sample (pre-training fill), only ever rises (stale reads by
        # the ingest thread are conservative)
        self._reader_floor = None
        self._floor_prev = None
        self.table = deque()           # (pos0, steps, outcome(np[P]))
        # guards table/eviction state: the Learner path extends from its
        # episode-feeder thread while the trainer thread publishes/samples
        self._table_lock = threading.Lock()
        self.total_added = 0
        self._pin = {}
        self._ingest = None
        self._ready = deque()
        self._ready_lock = threading.Lock()
        if device.type == 'cuda':
            self._ingest_stream = torch.cuda.Stream()
        if ingest_thread and device.type == 'cuda':
            self._ingest_q = queue_mod.Queue(maxsize=64)
            self._ingest = threading.Thread(target=self._ingest_loop,
                                            daemon=True)
            self._ingest.start()

    def _ingest_loop(self):
        import time
        torch.cuda.set_device(self.device)
        while True:
            episodes = self._ingest_q.get()
            try:
                for chunk in self._chunk(episodes):
                    # back-pressure: never write ring positions that alias
                    # rows at or above the consumer's reader floor (covers
                    # entries an in-flight captured gather may still read)
                    n_rows = sum(int(ep['steps']) for ep in chunk)
                    while True:
                        floor = self._reader_floor
                        if floor is None or \
                                self.head + n_rows - self.ring_T <= floor:
                            break
                        time.sleep(0.001)
                    with torch.cuda.stream(self._ingest_stream):
                        entries, n_rows = self._copy_block(chunk)
                        event = torch.cuda.Event()
                        event.record(self._ingest_stream)
                    with self._ready_lock:
                        self._ready.append((entries, n_rows, event))
            except Exception as e:   # noqa: BLE001 - never kill the ingester
                import sys
                print('replay ingest dropped a block: %r' % (e,),
                      file=sys.stderr, flush=True)
            finally:
                self._ingest_q.task_done()

    def _chunk(self, episodes):
        """Split an episode list so no block exceeds a quarter of the ring
        (bursts of long episodes can otherwise overflow a small ring)."""
        limit = max(512, self.ring_T // 4)
        chunk, rows = [], 0
        for ep in episodes:
            steps = int(ep['steps'])
            if steps > limit:
                continue                      # cannot ever fit: drop
            if chunk and rows + steps > limit:
                yield chunk
                chunk, rows = [], 0
            chunk.append(ep)
            rows += steps
        if chunk:
            yield chunk

    def publish(self, stream=None):
        """Make finished background/committed writes sampleable (consumer
        thread): extends the table after a stream-ordered wait on each
        write's completion event."""
        if self.device.type != 'cuda':
            return
        if stream is None:
            stream = torch.cuda.current_stream()
        with self._ready_lock:
            ready, self._ready = self._ready, deque()
        with self._table_lock:
            for entries, n_rows, event in ready:
                stream.wait_event(event)
                self.table.extend(entries)
                self.total_added += len(entries)
            self._evict()

    def flush(self):
        """Block until every queued episode block is published."""
        if self.device.type != 'cuda':
            return
        if self._ingest is not None:
            self._ingest_q.join()
        torch.cuda.synchronize()
        self.publish()

    def __len__(self):
        with self._table_lock:
            return len(self.table)

    def commit_traj(self, traj, g_rows, lens, outcomes, gate=None):
        """Copy finished device-recorded episodes (handyrl_amd/traj) into
        the ring DEVICE-TO-DEVICE and publish their table entries.

        g_rows/lens: int64[K] global trajectory rows and episode
        lengths; outcomes: float32[K, OUTCOME_P].  ``gate`` is an event
        recorded after the finishing worker's last service (its scatters
        to these rows) — by the pool protocol it has already fired, so
        the wait is free.  Returns the completion event (the caller must
        make the next trajectory write to these rows wait on it) or None
        when K == 0.  Runs on the ingest stream."""
        K = len(g_rows)
        if K == 0:
            return None
        lens = np.asarray(lens, dtype=np.int64)
        n = int(lens.sum())
        assert n <= self.ring_T // 4, 'episode burst larger than ring/4'
        import time as _time
        waited = 0.0
        while True:                    # reader-floor back-pressure
            floor = self._reader_floor
            if floor is None or self.head + n - self.ring_T <= floor:
                break
            _time.sleep(0.001)
            waited += 0.001
            if waited > 5.0:
                raise RuntimeError('replay ring writer starved (floor %r)'
                                   % (floor,))
        g_flat, t_flat = traj.episode_indices(np.asarray(g_rows), lens)
        dev = self.device
        head0 = self.head
        if dev.type == 'cuda':
            # `gate` is the finishing worker's LAST service event: by
            # protocol it has already fired (the worker only steps after
            # 'go', which follows the event sync), so waiting on it is
            # free — while gating on the whole main stream serialized the
            # commit behind every OTHER worker's in-flight forward
            # (measured as a 20% Geister regression)
            stream = self._ingest_stream if self._ingest is not None \
                else torch.cuda.current_stream()
            with torch.cuda.stream(stream):
                event = self._commit_copy(traj, g_flat, t_flat, head0, n,
                                          gate=gate)
        else:
            event = None
            self._commit_copy(traj, g_flat, t_flat, head0, n)
        entries, pos = [], head0
        for k in range(K):
            entries.append((pos, int(lens[k]),
                            np.asarray(outcomes[k], dtype=np.float32)))
            pos += int(lens[k])
        self.head = pos
        self.head_planned = max(self.head_planned, pos)
        with self._table_lock:
            if event is None:
                self.table.extend(entries)
                self.total_added += len(entries)
                self._evict()
        if event is not None:
            with self._ready_lock:
                self._ready.append((entries, n, event))
        return event

    def _stage_idx(self, key, arr):
        """Pinned staging for commit index arrays: async H2D made SAFE
        (the pinned buffer persists; a refill waits on the previous
        copy's event).  Runs inside the ingest-stream context.  Temporary
        pageable sources must never feed a non-blocking copy — a freed
        temporary under a late-reading hipMemcpyAsync supplies garbage
        INDICES to the gather below (HSAIL-fault class)."""
        if self.device.type != 'cuda':
            return torch.from_numpy(arr).to(self.device)
        n = arr.shape[0]
        pin, ev = self._pin.get(key, (None, None))
        if pin is None or pin.shape[0] < n:
            pin = torch.empty(max(int(n * 1.5), 4096), dtype=torch.int64,
                              pin_memory=True)
            ev = None
        if ev is not None:
            ev.synchronize()
        pin.numpy()[:n] = arr
        dev_t = pin[:n].to(self.device, non_blocking=True)
        ev = torch.cuda.Event()
        ev.record()
        self._pin[key] = (pin, ev)
        return dev_t

    # -- write path --------------------------------------------------------
    def _stage(self, key, parts, n, dtype, tail_shape):
        """Fill a pinned staging buffer from per-episode arrays (one memcpy
        each, GIL released) and return its device copy (async H2D).

        The previous block's H2D from this buffer may still be in flight
        (the ingest stream's copies queue behind the actor pool's DMA
        traffic under load), so refilling must wait on the event recorded
        after that copy — without it the in-flight DMA reads torn bytes
        and poisons the ring with garbage floats (NaN losses; diagnosed
        in profiles/learn_race*.log)."""
        if self.device.type != 'cuda':
            return torch.from_numpy(np.concatenate(parts)).to(self.device)
        pin, ev = self._pin.get(key, (None, None))
        if pin is None or pin.shape[0] < n:
            cap = max(int(n * 1.5), 1024)
            pin = torch.empty((cap,) + tail_shape, dtype=dtype, pin_memory=True)
            ev = None
        if ev is not None:
            ev.synchronize()
        view = pin.numpy()
        off = 0
        for part in parts:
            view[off:off + part.shape[0]] = part
            off += part.shape[0]
        dev = pin[:n].to(self.device, non_blocking=True)
        ev = torch.cuda.Event()
        ev.record()                    # ingest stream inside the ingest loop
        self._pin[key] = (pin, ev)
        return dev

    def _write_ring(self, dst, src):
        n = src.shape[0]
        p = self.head % self.ring_T
        first = min(n, self.ring_T - p)
        dst[p:p + first] = src[:first]
        if n > first:
            dst[:n - first] = src[first:]

    def _copy_block(self, episodes):
        """Ring-write one block of episodes; returns (table entries, rows)."""
        n = sum(int(ep['steps']) for ep in episodes)
        if n > self.ring_T:
            raise ValueError('episode block larger than the replay ring')
        for name, (dt, tail, key) in self.COLUMNS.items():
            self._write_ring(getattr(self, name), self._stage(
                name, [ep[key or name] for ep in episodes], n, dt, tail))
        entries = []
        pos = self.head
        for ep in episodes:
            oc = np.array([ep['outcome'][p] for p in range(self.OUTCOME_P)],
                          dtype=np.float32)
            entries.append((pos, int(ep['steps']), oc))
            pos += int(ep['steps'])
        self.head = pos
        return entries, n

    def _pick_window(self, m, fs):
        """One recency-biased episode pick + window cut (reference
        train.py:291-315 accept-loop semantics)."""
        while True:
            idx = random.randrange(m)
            accept = 1 - (m - 1 - idx) / m
            if random.random() < accept:
                break
        with self._table_lock:
            if not self.table:
                raise RuntimeError('replay table emptied during sampling')
            p0, steps, oc = self.table[min(idx, len(self.table) - 1)]
        train_st = random.randrange(1 + max(0, steps - fs))
        ed = min(train_st + fs, steps)
        return p0, steps, oc, train_st, ed

    def _update_reader_floor(self, sampled_min=None):
        # reader floor covers this fill AND the previous one (the previous
        # graph replay may still be in flight when this one is sampled).
        # `sampled_min` is the smallest episode base position actually
        # drawn this fill: the feeder thread may evict (and later overwrite)
        # the table head between our picks and this call, so table[0] alone
        # can fail to cover an entry that is already in the in-flight gather.
        with self._table_lock:
            if not self.table:
                floor_now = sampled_min
            else:
                floor_now = self.table[0][0]
                if sampled_min is not None:
                    floor_now = min(floor_now, sampled_min)
        if floor_now is None:
            return
        self._reader_floor = floor_now if self._floor_prev is None \
            else min(floor_now, self._floor_prev)
        self._floor_prev = floor_now

    def _evict(self):
        # With background ingest, evict against the PLANNED write frontier
        # (every queued-but-unwritten row counted): entries stay at least
        # EVICT_MARGIN rows clear of any ring position an in-flight or
        # queued block can touch, so concurrent ingest writes never alias
        # rows a captured gather may still read.  self.head alone lags the
        # queue and is advanced by the ingest thread (racy to read here).
        if self._ingest is not None:
            min_valid = self.head_planned - self.ring_T + EVICT_MARGIN
        else:
            min_valid = self.head - self.ring_T
        max_eps = self.args['maximum_episodes']
        while self.table and (self.table[0][0] < min_valid or
                              len(self.table) > max_eps):
            self.table.popleft()

    def extend(self, episodes):
        """Append columnar episodes (numpy fields) to the device ring.
        Episode stubs already committed device-side (traj mode,
        ep['committed']) pass through untouched — their rows are in the
        ring already."""
        episodes = [ep for ep in episodes if not ep.get('committed')]
        if not episodes:
            return
        limit = max(512, self.ring_T // 4)
        self.head_planned += sum(int(ep['steps']) for ep in episodes
                                 if int(ep['steps']) <= limit)
        if self._ingest is not None:
            self._ingest_q.put(episodes)
            with self._table_lock:
                self._evict()          # clear the planned frontier's alias zone
            return
        for chunk in self._chunk(episodes):
            entries, _ = self._copy_block(chunk)
            with self._table_lock:
                self.table.extend(entries)
                self.total_added += len(entries)
        with self._table_lock:
            self._evict()

    def trim(self, maximum):
        with self._table_lock:
            while len(self.table) > maximum:
                self.table.popleft()


class DeviceReplay(ColumnRingReplay):
    """Solo feed-forward replay (the flagship Hungry Geese path): rows
    store CANONICAL per-game obs + per-seat alive/action/prob/value; the
    per-seat channel rotation happens inside gather_batch on device."""

    COLUMNS = {
        'obs': (torch.uint8, OBS_SHAPE, None),
        'alive': (torch.bool, (4,), None),
        'action': (torch.int32, (4,), None),
        'prob': (torch.float32, (4,), None),
        'value': (torch.float32, (4,), None),
    }
    OUTCOME_P = 4

    def __init__(self, args, device, bytes_budget=4 << 30, ingest_thread=False):
        assert args.get('burn_in_steps', 0) == 0, \
            'DeviceReplay supports feed-forward (no burn-in) training'
        super().__init__(args, device, bytes_budget, ingest_thread)
        from .envs.vec_geese import CHMAP
        self._chmap = torch.from_numpy(CHMAP).to(device)     # (4, 17)
        self._arange_cache = {}

    # -- device-side ingest (traj mode) --------------------------------------
    def _commit_copy(self, traj, g_flat, t_flat, head0, n, gate=None):
        dev = self.device
        # stage the index H2Ds BEFORE gating on the main stream: gating
        # them too makes the next commit's pinned-buffer reuse sync wait
        # behind every in-flight forward (measured as the short-episode
        # regime's 2x service cost); only the ring GATHERS below must
        # order after the main stream's trajectory scatters
        g_t = self._stage_idx('commit_g', g_flat)
        t_t = self._stage_idx('commit_t', t_flat)
        dst = (torch.arange(n, device=dev, dtype=torch.int64) + head0) \
            % self.ring_T
        if gate is not None:
            torch.cuda.current_stream().wait_event(gate)
        self.obs[dst] = traj.obs[g_t, t_t]
        self.alive[dst] = traj.alive[g_t, t_t]
        rec = traj.rec[g_t, t_t]                     # (n, 4, 3)
        self.action[dst] = rec[..., 0].to(torch.int32)
        self.prob[dst] = rec[..., 1]
        self.value[dst] = rec[..., 2]
        if dev.type != 'cuda':
            return None
        event = torch.cuda.Event()
        event.record()
        return event

    # -- sample path ---------------------------------------------------------
    def sample_indices(self, batch_size):
        """Recency-biased episode picks + window cuts + solo-seat choice
        (reference train.py:291-315 / make_batch solo semantics).
        Returns int64/float32 numpy arrays of length B."""
        args = self.args
        fs = args['forward_steps']
        n = len(self.table)
        assert n > 0, 'empty replay'
        pos0 = np.empty(batch_size, dtype=np.int64)
        start = np.empty(batch_size, dtype=np.int64)
        length = np.empty(batch_size, dtype=np.int64)
        seat = np.empty(batch_size, dtype=np.int64)
        outcome = np.empty((batch_size, 4), dtype=np.float32)
        inv_total = np.empty(batch_size, dtype=np.float32)
        m = min(n, args['maximum_episodes'])
        for b in range(batch_size):
            p0, steps, oc, train_st, ed = self._pick_window(m, fs)
            pos0[b] = p0 + train_st
            start[b] = train_st
            length[b] = ed - train_st
            seat[b] = random.randrange(4)
            outcome[b] = oc
            inv_total[b] = 1.0 / steps
        self._update_reader_floor(int((pos0 - start).min()))
        return pos0, start, length, seat, outcome, inv_total

    def gather_batch(self, pos0, start, length, seat, outcome, inv_total):
        """Build the (B, T, 1, ...) training batch on device.  All inputs
        are device tensors of length B; every op here is hipGraph-capturable
        (per-step variability flows through these index tensors)."""
        args = self.args
        B = pos0.shape[0]
        T = args['forward_steps']
        dev = self.device
        t_range = torch.arange(T, device=dev)
        rows = (pos0.unsqueeze(1) + t_range) % self.ring_T      # (B, T)
        in_range = (t_range.unsqueeze(0) < length.unsqueeze(1))  # (B, T) bool

        flat = rows.reshape(-1)
        sel = seat.unsqueeze(1).expand(B, T).reshape(-1)

        # canonical row gather + per-seat channel rotation (CHMAP)
        obs_c = self.obs[flat].reshape(-1, 17, 77)               # (B*T, 17, 77)
        chmap_bt = self._chmap[sel]                              # (B*T, 17)
        nbt = obs_c.shape[0]
        ar = self._arange_cache.get(nbt)
        if ar is None:
            ar = torch.arange(nbt, device=dev).unsqueeze(1)
            self._arange_cache[nbt] = ar
        obs = obs_c[ar, chmap_bt].reshape(-1, 17, 7, 11)         # (B*T, 17,7,11)
        alive = self.alive[flat, sel]
        act = self.action[flat, sel].long()
        prob = self.prob[flat, sel]
        val = self.value[flat, sel]

        in_r = in_range.reshape(-1)
        tmask = (alive & in_r).float()
        obs = obs * tmask.to(torch.uint8).view(-1, 1, 1, 1)
        oc_b = outcome.gather(1, seat.unsqueeze(1)).squeeze(1)   # (B,)
        oc_bt = oc_b.unsqueeze(1).expand(B, T).reshape(-1)
        v = torch.where(in_r, val * tmask, oc_bt)
        prob = torch.where(tmask.bool(), prob, torch.ones_like(prob))
        act = act * tmask.long()
        amask = torch.where(tmask.bool(), torch.zeros_like(prob),
                            torch.full_like(prob, 1e32))
        amask4 = amask.unsqueeze(1).expand(B * T, 4).contiguous()
        progress = torch.where(
            in_r,
            (start.unsqueeze(1) + t_range).reshape(-1).float() *
            inv_total.unsqueeze(1).expand(B, T).reshape(-1),
            torch.ones(1, device=dev))

        def shp(x, *tail):
            return x.reshape(B, T, 1, *tail)

        zeros = torch.zeros(B, T, 1, 1, device=dev)
        return {
            'observation': shp(obs, 17, 7, 11),
            'selected_prob': shp(prob, 1),
            'value': shp(v, 1),
            'action': shp(act, 1),
            'outcome': oc_b.view(B, 1, 1, 1),
            'reward': zeros,
            'return': zeros.clone(),
            'episode_mask': in_range.float().view(B, T, 1, 1),
            'turn_mask': shp(tmask, 1),
            'observation_mask': shp(tmask.clone(), 1),
            'action_mask': shp(amask4, 4),
            'progress': progress.view(B, T, 1),
        }


class TurnDeviceReplay(ColumnRingReplay):
    """Turn-based (observation=False) replay for recurrent envs (Geister):
    rows store the MOVER's observation/legality plus per-player value-side
    columns; gather_batch reproduces make_batch's turn-based columnar
    fields on device (parity-tested in tests/test_turn_replay.py;
    reference semantics: replay window cut train.py:291-315, batch
    padding train.py:92-110, turn seat selection train.py:65-68).

    Episodes are the columnar turn-based format produced by the Geister
    actor pools (actor_geister.py)."""

    COLUMNS = {
        'scalar': (torch.uint8, (18,), None),
        'board': (torch.uint8, (7, 6, 6), None),
        'mask': (torch.bool, (214,), None),          # legality bools
        'turn': (torch.int8, (), None),
        'action': (torch.int16, (), None),
        'prob': (torch.float32, (), None),
        'value': (torch.float32, (), None),
        'reward': (torch.float32, (2,), None),
        'ret': (torch.float32, (2,), 'return'),
    }
    OUTCOME_P = 2

    def _commit_copy(self, traj, g_flat, t_flat, head0, n, gate=None):
        """Turn-based commit (GeisterTrajRecorder rings -> ring columns).
        reward is the constant -0.01 per step (geister.py reward());
        the discounted-return column is computed in closed form:
        ret[t] = -0.01 * (1 - gamma^(S-t)) / (1 - gamma) — identical math
        to the worker's backward scan."""
        dev = self.device
        g_t = self._stage_idx('commit_g', g_flat)
        t_t = self._stage_idx('commit_t', t_flat)
        # steps-to-go per flat row: S - t (episode boundaries at t == 0)
        starts = np.flatnonzero(t_flat == 0)
        lens_ep = np.diff(np.append(starts, len(t_flat)))
        k_np = (np.repeat(lens_ep, lens_ep) - t_flat).astype(np.int64)
        k_t = self._stage_idx('commit_k', k_np)
        dst = (torch.arange(n, device=dev, dtype=torch.int64) + head0) \
            % self.ring_T
        if gate is not None and dev.type == 'cuda':
            torch.cuda.current_stream().wait_event(gate)
        self.scalar[dst] = traj.scalar[g_t, t_t]
        self.board[dst] = traj.board[g_t, t_t]
        self.mask[dst] = traj.mask[g_t, t_t]
        self.turn[dst] = traj.turn[g_t, t_t]
        apv = traj.apv[g_t, t_t]                      # (n, 3)
        self.action[dst] = apv[:, 0].to(torch.int16)
        self.prob[dst] = apv[:, 1]
        self.value[dst] = apv[:, 2]
        gamma = float(self.args['gamma'])
        ret = (-0.01) * (1.0 - torch.pow(
            torch.full((n,), gamma, device=dev), k_t.float())) / (1.0 - gamma)
        self.reward[dst] = torch.full((n, 2), -0.01, device=dev)
        self.ret[dst] = ret.unsqueeze(1).expand(n, 2)
        if dev.type != 'cuda':
            return None
        event = torch.cuda.Event()
        event.record()
        return event

    def sample_indices(self, batch_size):
        """Recency-biased picks + window cuts (no seat choice: the mover
        axis is decided per step by the recorded turn column).  With
        burn_in_steps > 0 the window is extended backwards by up to
        burn_in rows and a per-sample ``lead`` pad (the shortfall at the
        episode start) is returned as a sixth array."""
        args = self.args
        fs = args['forward_steps']
        burn_in = args.get('burn_in_steps', 0)
        n = len(self.table)
        assert n > 0, 'empty replay'
        pos0 = np.empty(batch_size, dtype=np.int64)
        start = np.empty(batch_size, dtype=np.int64)
        length = np.empty(batch_size, dtype=np.int64)
        lead = np.empty(batch_size, dtype=np.int64)
        outcome = np.empty((batch_size, 2), dtype=np.float32)
        inv_total = np.empty(batch_size, dtype=np.float32)
        m = min(n, args['maximum_episodes'])
        for b in range(batch_size):
            p0, steps, oc, train_st, ed = self._pick_window(m, fs)
            st = max(0, train_st - burn_in)
            pos0[b] = p0 + st
            start[b] = st
            length[b] = ed - st
            lead[b] = burn_in - (train_st - st)
            outcome[b] = oc
            inv_total[b] = 1.0 / steps
        self._update_reader_floor(int((pos0 - start).min()))
        if burn_in == 0:
            return pos0, start, length, outcome, inv_total
        return pos0, start, length, outcome, inv_total, lead

    def gather_batch(self, pos0, start, length, outcome, inv_total,
                     lead=None):
        """Build the turn-based (B, T, P, ...) batch on device: obs/prob/
        action/action_mask on the mover axis (P=1), value-side fields on
        the player axis (P=2).  All inputs are device tensors of length B;
        every op is hipGraph-capturable.  Out-of-window rows read garbage
        ring memory, so every float path uses torch.where (never
        multiply-by-mask, which leaks NaN*0).

        With ``lead`` (burn_in configs) the batch spans burn_in +
        forward_steps rows: column t maps to window row t - lead[b], rows
        before the window (burn-in shortfall at the episode start) pad
        like make_batch's prefix padding (v -> 0, not outcome)."""
        if lead is not None:
            return self._gather_burnin(pos0, start, length, outcome,
                                       inv_total, lead)
        args = self.args
        B = pos0.shape[0]
        T = args['forward_steps']
        dev = self.device
        t_range = torch.arange(T, device=dev)
        rows = (pos0.unsqueeze(1) + t_range) % self.ring_T      # (B, T)
        in_range = t_range.unsqueeze(0) < length.unsqueeze(1)   # (B, T)
        flat = rows.reshape(-1)
        in_r = in_range.reshape(-1)                             # (B*T,)

        scalar = self.scalar[flat].float() * in_r.unsqueeze(1)  # u8: mul safe
        board = self.board[flat].float() * in_r.view(-1, 1, 1, 1)
        legal = self.mask[flat] & in_r.unsqueeze(1)
        amask = torch.where(legal, torch.zeros((), device=dev),
                            torch.full((), 1e32, device=dev))
        turn = self.turn[flat].long().clamp(0, 1)               # (B*T,)
        act = (self.action[flat].long() * in_r.long())
        prob = torch.where(in_r, self.prob[flat],
                           torch.ones((), device=dev))
        oc_bt = outcome.unsqueeze(1).expand(B, T, 2).reshape(-1, 2)
        tmask = torch.zeros(B * T, 2, device=dev)
        tmask.scatter_(1, turn.unsqueeze(1), in_r.float().unsqueeze(1))
        v_m = torch.zeros(B * T, 2, device=dev)
        v_m.scatter_(1, turn.unsqueeze(1),
                     torch.where(in_r, self.value[flat],
                                 torch.zeros((), device=dev)).unsqueeze(1))
        v = torch.where(in_r.unsqueeze(1), v_m, oc_bt)
        rew = torch.where(in_r.unsqueeze(1), self.reward[flat],
                          torch.zeros((), device=dev))
        ret = torch.where(in_r.unsqueeze(1), self.ret[flat],
                          torch.zeros((), device=dev))
        progress = torch.where(
            in_r,
            (start.unsqueeze(1) + t_range).reshape(-1).float() *
            inv_total.unsqueeze(1).expand(B, T).reshape(-1),
            torch.ones((), device=dev))

        def mover(x, *tail):                                    # (B, T, 1, ...)
            return x.reshape(B, T, 1, *tail)

        return {
            'observation': {'scalar': mover(scalar, 18),
                            'board': mover(board, 7, 6, 6)},
            'selected_prob': mover(prob, 1),
            'value': v.reshape(B, T, 2, 1),
            'action': mover(act, 1),
            'outcome': outcome.view(B, 1, 2, 1),
            'reward': rew.reshape(B, T, 2, 1),
            'return': ret.reshape(B, T, 2, 1),
            'episode_mask': in_range.float().view(B, T, 1, 1),
            'turn_mask': tmask.reshape(B, T, 2, 1),
            'observation_mask': tmask.reshape(B, T, 2, 1).clone(),
            'action_mask': mover(amask, 214),
            'progress': progress.view(B, T, 1),
        }

    def _gather_burnin(self, pos0, start, length, outcome, inv_total, lead):
        """Burn-in variant of gather_batch: T = burn_in + forward_steps,
        with a per-sample leading pad."""
        args = self.args
        B = pos0.shape[0]
        T = args['burn_in_steps'] + args['forward_steps']
        dev = self.device
        t_range = torch.arange(T, device=dev)
        k = t_range.unsqueeze(0) - lead.unsqueeze(1)            # (B, T)
        valid = (k >= 0) & (k < length.unsqueeze(1))
        after = k >= length.unsqueeze(1)                        # tail pad
        rows = (pos0.unsqueeze(1) + k.clamp(min=0)) % self.ring_T
        flat = rows.reshape(-1)
        in_r = valid.reshape(-1)                                # (B*T,)
        aft = after.reshape(-1)

        scalar = self.scalar[flat].float() * in_r.unsqueeze(1)
        board = self.board[flat].float() * in_r.view(-1, 1, 1, 1)
        legal = self.mask[flat] & in_r.unsqueeze(1)
        amask = torch.where(legal, torch.zeros((), device=dev),
                            torch.full((), 1e32, device=dev))
        turn = self.turn[flat].long().clamp(0, 1)
        act = (self.action[flat].long() * in_r.long())
        prob = torch.where(in_r, self.prob[flat],
                           torch.ones((), device=dev))
        oc_bt = outcome.unsqueeze(1).expand(B, T, 2).reshape(-1, 2)
        tmask = torch.zeros(B * T, 2, device=dev)
        tmask.scatter_(1, turn.unsqueeze(1), in_r.float().unsqueeze(1))
        v_m = torch.zeros(B * T, 2, device=dev)
        v_m.scatter_(1, turn.unsqueeze(1),
                     torch.where(in_r, self.value[flat],
                                 torch.zeros((), device=dev)).unsqueeze(1))
        # prefix pads stay 0; only rows PAST the episode splice the outcome
        v = torch.where(in_r.unsqueeze(1), v_m,
                        torch.where(aft.unsqueeze(1), oc_bt,
                                    torch.zeros((), device=dev)))
        rew = torch.where(in_r.unsqueeze(1), self.reward[flat],
                          torch.zeros((), device=dev))
        ret = torch.where(in_r.unsqueeze(1), self.ret[flat],
                          torch.zeros((), device=dev))
        progress = torch.where(
            in_r,
            (start.unsqueeze(1) + k).reshape(-1).float() *
            inv_total.unsqueeze(1).expand(B, T).reshape(-1),
            torch.ones((), device=dev))

        def mover(x, *tail):                                    # (B, T, 1, ...)
            return x.reshape(B, T, 1, *tail)

        return {
            'observation': {'scalar': mover(scalar, 18),
                            'board': mover(board, 7, 6, 6)},
            'selected_prob': mover(prob, 1),
            'value': v.reshape(B, T, 2, 1),
            'action': mover(act, 1),
            'outcome': outcome.view(B, 1, 2, 1),
            'reward': rew.reshape(B, T, 2, 1),
            'return': ret.reshape(B, T, 2, 1),
            'episode_mask': valid.float().view(B, T, 1, 1),
            'turn_mask': tmask.reshape(B, T, 2, 1),
            'observation_mask': tmask.reshape(B, T, 2, 1).clone(),
            'action_mask': mover(amask, 214),
            'progress': progress.view(B, T, 1),
        }
