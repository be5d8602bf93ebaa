"""hipGraph capture for the launch-bound hot loops.

The learner step (a ~300-kernel chain of small convs, elementwise loss
math, fused scans and the Adam update at these model sizes) and the actor
forward are dominated by per-launch gaps on MI355X, not kernel time
(profiles/bench_kernel_stats_r01.txt) — so both are captured once into a
hipGraph and replayed, with inputs staged through static device buffers.

Constraints honored here:
* no host syncs inside capture (compose_losses keeps dcnt as a tensor);
* Adam runs in capturable mode (device-side step counters);
* static shapes (the bench batch is fixed; actor batches are padded to
  fixed buckets by GeeseActorPool);
* the RCCL gradient all-reduce is captured too (graph-capturable on ROCm);
  if capture fails anywhere the caller falls back to the eager path.
"""

import threading

import torch
import torch.nn as nn

from .util import map_r, bimap_r

# Graph CAPTURE must never overlap a replay from another thread: torch's
# graph-safe RNG is process-global, and a replay issued mid-capture fails
# with "Offset increment outside graph capture" (seen when the Learner's
# GPU-actor thread replayed service graphs while the trainer thread
# captured its train step).  Capture sites hold this lock; so does the
# Learner's actor loop around its replay bursts.
CAPTURE_LOCK = threading.RLock()


def apply_grad_guard(params, counter=None):
    from .train import apply_grad_guard as _g
    _g(params, counter)


def _copy_into(static, src, non_blocking=True):
    bimap_r(static, src, lambda dst, s: dst.copy_(s, non_blocking=non_blocking))


class GraphedTrainStep:
    """Whole-train-step capture: forward + loss + backward + all-reduce +
    grad-clip + Adam, replayed per step."""

    def __init__(self, trainer, example_batch_cpu, warmup_iters=3):
        from .train import compute_loss
        self._compute_loss = compute_loss
        self.trainer = trainer
        assert trainer.device.type == 'cuda'
        assert trainer.wrapped_model.init_hidden([1, 1]) is None, \
            'graphed train step supports feed-forward models only'

        self.static = map_r(example_batch_cpu,
                            lambda t: t.to(trainer.device).clone())

        for group in trainer.optimizer.param_groups:
            group['capturable'] = True

        trainer.model.train()
        stream = torch.cuda.Stream()
        stream.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(stream):
            for _ in range(warmup_iters):
                self._run()
        torch.cuda.current_stream().wait_stream(stream)

        self.graph = torch.cuda.CUDAGraph()
        with CAPTURE_LOCK, torch.cuda.graph(
                    self.graph, capture_error_mode='thread_local'):
            self.losses, self.dcnt = self._run()

    def _run(self):
        tr = self.trainer
        if tr.use_amp:
            with torch.autocast('cuda', dtype=torch.bfloat16):
                losses, dcnt = self._compute_loss(self.static, tr.wrapped_model,
                                                  None, tr.args)
        else:
            losses, dcnt = self._compute_loss(self.static, tr.wrapped_model,
                                              None, tr.args)
        tr.optimizer.zero_grad(set_to_none=False)
        losses['total'].backward()
        if tr.grad_guard:
            # capturable finite-guard + activation counter: one spike step
            # (inf grads with a still-finite loss) otherwise turns the clip
            # scale into NaN and poisons the weights permanently
            # (BASELINE.md diagnosis); fires are reported per epoch
            apply_grad_guard(tr.params, tr.guard_fires)
        tr.reducer.allreduce_()
        nn.utils.clip_grad_norm_(tr.params, 4.0)
        tr.optimizer.step()
        return losses, dcnt

    def step(self, batch_cpu):
        """Copy the CPU batch into the static buffers and replay.

        Returns (losses, dcnt) as DEVICE tensors that alias the graph's
        static outputs — read them (.item()) before the next replay."""
        _copy_into(self.static, batch_cpu)
        self.graph.replay()
        self.trainer.steps += 1
        return self.losses, self.dcnt


class GraphedActorForward:
    """Per-bucket capture of {uint8 obs -> bf16 forward -> masked softmax
    sample -> packed (action, prob, value)} for the GPU actor pool."""

    def __init__(self, model, device, warmup_iters=2, fused=None,
                 canonical=True, traj=None):
        from . import ops
        from .envs.vec_geese import CHMAP
        self._ops = ops
        self.model = model
        self.device = device
        self.fused = fused        # GeeseFusedEval: hand-written MFMA path
        self.canonical = canonical  # obs rows are per GAME; out rows x4
        self.traj = traj          # TrajRecorder: in-graph HBM recording
        self._chmap = torch.from_numpy(CHMAP).to(device)
        self.graphs = {}

    def _fwd_sample(self, obs_u8, zero_mask):
        with torch.no_grad():
            if self.canonical:
                if self.fused is not None:
                    out = self.fused.forward_canonical(obs_u8)
                else:
                    G = obs_u8.shape[0]
                    obs_f = obs_u8.reshape(G, 17, 77).float()
                    obs4 = obs_f[:, self._chmap].reshape(G * 4, 17, 7, 11)
                    with torch.autocast('cuda', dtype=torch.bfloat16):
                        out = self.model(obs4, None)
            elif self.fused is not None:
                out = self.fused.forward(obs_u8)
            else:
                obs_f = obs_u8.float()
                with torch.autocast('cuda', dtype=torch.bfloat16):
                    out = self.model(obs_f, None)
            policy = out['policy'].float()
            value = out['value'].float()
            uniform = torch.rand(policy.shape[0], device=self.device)
            actions, probs = self._ops.masked_sample(policy, zero_mask, uniform)
            return torch.cat([actions.float().unsqueeze(1),
                              probs.unsqueeze(1), value], dim=1)

    def _capture(self, bucket, n_actions):
        was_training = self.model.training
        self.model.eval()
        static_obs = torch.zeros(bucket, 17, 7, 11, dtype=torch.uint8,
                                 device=self.device)
        rows = bucket * 4 if self.canonical else bucket
        zero_mask = torch.zeros(rows, n_actions, device=self.device)
        static_idx = None
        if self.traj is not None:
            # in-graph HBM trajectory recording: game-row / step-index
            # inputs (padded bucket rows point at the scratch row)
            static_idx = (torch.full((bucket,), self.traj.scratch_row,
                                     dtype=torch.int64, device=self.device),
                          torch.zeros(bucket, dtype=torch.int64,
                                      device=self.device))

        def run_once():
            packed = self._fwd_sample(static_obs, zero_mask)
            if self.traj is not None:
                self.traj.record_(static_obs, packed,
                                  static_idx[0], static_idx[1])
            return packed

        stream = torch.cuda.Stream()
        stream.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(stream):
            for _ in range(2):
                run_once()
        torch.cuda.current_stream().wait_stream(stream)
        graph = torch.cuda.CUDAGraph()
        with CAPTURE_LOCK, torch.cuda.graph(
                graph, capture_error_mode='thread_local'):
            packed = run_once()
        if was_training:
            self.model.train()
        # CRITICAL: every tensor a captured graph reads must stay
        # python-referenced — the graph holds raw device pointers, and a
        # freed block gets reused by later allocations (the train-graph
        # capture reliably recycled a freed index/mask tensor here, turning
        # replays into OOB scatters / garbage sampling masks)
        self.graphs[bucket] = (graph, static_obs, packed, static_idx,
                               zero_mask)

    def run(self, obs_u8_cpu, n_actions=4):
        """obs_u8_cpu: torch uint8 tensor (M, 17, 7, 11) on CPU (M <= bucket
        after caller padding decides the bucket). Returns packed (M, 3) on
        device."""
        M = obs_u8_cpu.shape[0]
        bucket = self._bucket(M)
        if bucket not in self.graphs:
            self._capture(bucket, n_actions)
        graph, static_obs, packed, _idx, _zm = self.graphs[bucket]
        static_obs[:M].copy_(obs_u8_cpu, non_blocking=True)
        # rows [M:bucket) keep stale data; every op is row-independent and
        # the outputs are sliced to the live rows
        graph.replay()
        return packed[:M * 4] if self.canonical else packed[:M]

    def _bucket(self, M):
        # canonical: M counts GAMES (4 net rows each); 32-game buckets =
        # 128-row forwards (the fused MFMA kernels take any row count —
        # the 256-row multiple was only ever a MIOpen find-cache guard)
        q = 32 if self.canonical else 256
        return q * ((M + q - 1) // q)

    def capture_service_core(self, gidx_const, tidx_dev, M, n_actions=4):
        """Service CORE capture: fused forward + masked sample + in-graph
        trajectory scatter + device step-counter update, with the pinned
        H2D/D2H left to the caller (captured host-memcpy nodes faulted on
        hardware — HSA exception at replay — so the copies stay eager:
        still only ~4 host calls per service).

        Returns (graph, static_obs, packed)."""
        bucket = self._bucket(M)
        static_obs = torch.zeros(bucket, 17, 7, 11, dtype=torch.uint8,
                                 device=self.device)
        rows = bucket * 4 if self.canonical else bucket
        zero_mask = torch.zeros(rows, n_actions, device=self.device)
        max_t = self.traj.max_steps if self.traj is not None else 1

        def run_once():
            packed = self._fwd_sample(static_obs, zero_mask)
            if self.traj is not None:
                self.traj.record_(static_obs, packed, gidx_const, tidx_dev)
                tidx_dev.add_(1).clamp_(0, max_t - 1)
            return packed

        was_training = self.model.training
        self.model.eval()
        stream = torch.cuda.Stream()
        stream.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(stream):
            for _ in range(2):
                run_once()
        torch.cuda.current_stream().wait_stream(stream)
        graph = torch.cuda.CUDAGraph()
        with CAPTURE_LOCK, torch.cuda.graph(
                graph, capture_error_mode='thread_local'):
            packed = run_once()
        if was_training:
            self.model.train()
        tidx_dev.zero_()          # warmup replays polluted the counters
        torch.cuda.synchronize()
        # zero_mask rides along: the graph reads it every replay
        return graph, static_obs, packed, zero_mask

    def capture_service(self, obs_src, out_dst, gidx_const, tidx_dev, M,
                        n_actions=4):
        """Whole-service capture for one (worker, slot) of the multiproc
        actor pool: ONE replay executes H2D of the observations (from a
        pinned staging tensor or a hipHostRegister'd shm view), the fused
        forward + masked sample, the in-graph HBM trajectory scatter with
        a device-resident step counter, and the D2H of the packed results.

        gidx_const: (bucket,) int64 device — global trajectory rows
        (bucket padding points at the scratch row); tidx_dev: (bucket,)
        int64 device step counters, incremented in-graph and clamped (the
        caller zeroes finished games' entries before the next replay).
        Returns the graph; replaying it IS the service."""
        bucket = self._bucket(M)
        static_obs = torch.zeros(bucket, 17, 7, 11, dtype=torch.uint8,
                                 device=self.device)
        rows = bucket * 4 if self.canonical else bucket
        zero_mask = torch.zeros(rows, n_actions, device=self.device)
        R = M * 4 if self.canonical else M
        max_t = self.traj.max_steps if self.traj is not None else 1

        def run_once():
            static_obs[:M].copy_(obs_src[:M], non_blocking=True)
            packed = self._fwd_sample(static_obs, zero_mask)
            if self.traj is not None:
                self.traj.record_(static_obs, packed, gidx_const, tidx_dev)
                tidx_dev.add_(1).clamp_(0, max_t - 1)
            out_dst[:R].copy_(packed[:R], non_blocking=True)

        was_training = self.model.training
        self.model.eval()
        stream = torch.cuda.Stream()
        stream.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(stream):
            for _ in range(2):
                run_once()
        torch.cuda.current_stream().wait_stream(stream)
        graph = torch.cuda.CUDAGraph()
        with CAPTURE_LOCK, torch.cuda.graph(
                graph, capture_error_mode='thread_local'):
            run_once()
        if was_training:
            self.model.train()
        tidx_dev.zero_()          # warmup replays polluted the counters
        torch.cuda.synchronize()
        graph._keep_alive = (static_obs, zero_mask, obs_src, out_dst)
        return graph

    def run_async(self, obs_pinned, M, out_pinned, event, n_actions=4,
                  idx_pinned=None):
        """Pipelined variant: H2D from a pinned staging tensor, replay, and
        an async D2H of the packed result into ``out_pinned``; ``event``
        records completion.  No host sync — the caller overlaps CPU work
        and waits on the event.

        With in-graph trajectory recording, ``idx_pinned`` is a pinned
        (2, bucket) int64 tensor of (game row, step index) per obs row,
        padded rows pointing at the recorder's scratch row."""
        bucket = self._bucket(M)
        if bucket not in self.graphs:
            self._capture(bucket, n_actions)
        graph, static_obs, packed, static_idx, _zm = self.graphs[bucket]
        static_obs[:M].copy_(obs_pinned[:M], non_blocking=True)
        if static_idx is not None:
            static_idx[0].copy_(idx_pinned[0, :bucket], non_blocking=True)
            static_idx[1].copy_(idx_pinned[1, :bucket], non_blocking=True)
        graph.replay()
        R = M * 4 if self.canonical else M
        out_pinned[:R].copy_(packed[:R], non_blocking=True)
        event.record()


class GraphedReplayTrainStep:
    """Train-step capture with the batch GATHERED FROM DEVICE REPLAY inside
    the graph: one replay executes sample-gather + forward + loss +
    backward + all-reduce + grad-clip + Adam.  Per-step variability enters
    through six tiny index tensors (B elements each)."""

    def __init__(self, trainer, replay, batch_size, warmup_iters=3):
        from .train import compute_loss
        self._compute_loss = compute_loss
        self.trainer = trainer
        self.replay = replay
        self.batch_size = batch_size
        dev = trainer.device
        B = batch_size
        self.idx = {
            'pos0': torch.zeros(B, dtype=torch.int64, device=dev),
            'start': torch.zeros(B, dtype=torch.int64, device=dev),
            'length': torch.zeros(B, dtype=torch.int64, device=dev),
            'seat': torch.zeros(B, dtype=torch.int64, device=dev),
            'outcome': torch.zeros(B, 4, device=dev),
            'inv_total': torch.zeros(B, device=dev),
        }
        trainer.model.train()
        self.graph = None
        if dev.type != 'cuda':
            return                    # eager path (CPU tests)
        for group in trainer.optimizer.param_groups:
            group['capturable'] = True

        stream = torch.cuda.Stream()
        stream.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(stream):
            for _ in range(warmup_iters):
                self._fill()
                self._run()
        torch.cuda.current_stream().wait_stream(stream)

        try:
            self.graph = torch.cuda.CUDAGraph()
            self._fill()
            with CAPTURE_LOCK, torch.cuda.graph(
                    self.graph, capture_error_mode='thread_local'):
                self.losses, self.dcnt = self._run()
        except Exception as e:      # noqa: BLE001 - run eager if capture fails
            import sys
            print('replay train-step capture failed, running eager: %r' % (e,),
                  file=sys.stderr)
            self.graph = None

    def _fill(self):
        self.replay.publish()          # admit finished background ingests
        pos0, start, length, seat, outcome, inv_total = \
            self.replay.sample_indices(self.batch_size)
        # keep the host arrays referenced until the next fill: the async
        # H2D below reads them after this function returns (and the stash
        # doubles as diagnosis ground truth — see tools/learning_check.py)
        self._last_host_idx = {
            'pos0': pos0, 'start': start, 'length': length, 'seat': seat,
            'outcome': outcome, 'inv_total': inv_total}
        for key, arr in (('pos0', pos0), ('start', start), ('length', length),
                         ('seat', seat), ('outcome', outcome),
                         ('inv_total', inv_total)):
            # blocking H2D: `arr` is a temporary pageable numpy array; an
            # async copy that outlives it feeds garbage SAMPLE INDICES to
            # the captured ring gather (intermittent batch corruption)
            self.idx[key].copy_(torch.from_numpy(arr))

    def _run(self):
        tr = self.trainer
        batch = self.replay.gather_batch(
            self.idx['pos0'], self.idx['start'], self.idx['length'],
            self.idx['seat'], self.idx['outcome'], self.idx['inv_total'])
        if tr.use_amp:
            with torch.autocast('cuda', dtype=torch.bfloat16):
                losses, dcnt = self._compute_loss(batch, tr.wrapped_model,
                                                  None, tr.args)
        else:
            losses, dcnt = self._compute_loss(batch, tr.wrapped_model,
                                              None, tr.args)
        tr.optimizer.zero_grad(set_to_none=False)
        losses['total'].backward()
        if tr.grad_guard:
            # capturable finite-guard + activation counter: one spike step
            # (inf grads with a still-finite loss) otherwise turns the clip
            # scale into NaN and poisons the weights permanently
            # (BASELINE.md diagnosis); fires are reported per epoch
            apply_grad_guard(tr.params, tr.guard_fires)
        tr.reducer.allreduce_()
        nn.utils.clip_grad_norm_(tr.params, 4.0)
        tr.optimizer.step()
        return losses, dcnt

    def step(self):
        """Sample indices on the host, copy them in, replay the graph."""
        self._fill()
        self.trainer.steps += 1
        if self.graph is not None:
            self.graph.replay()
            return self.losses, self.dcnt
        return self._run()


class GraphedRecurrentTrainStep:
    """Turn-based/recurrent counterpart of GraphedReplayTrainStep: the
    batch is gathered from a TurnDeviceReplay ring inside the step, the
    DRC forward runs the per-timestep RNN loop (train.forward_prediction)
    from a STATIC zero initial hidden, and on GPU the whole step attempts
    hipGraph capture (eager fallback otherwise — the capture of the
    T-step ConvLSTM loop is exercised on hardware, not assumed)."""

    def __init__(self, trainer, replay, batch_size, n_players=2,
                 warmup_iters=2):
        from .train import compute_loss
        self._compute_loss = compute_loss
        self.trainer = trainer
        self.replay = replay
        self.batch_size = batch_size
        dev = trainer.device
        B = batch_size
        self.idx = {
            'pos0': torch.zeros(B, dtype=torch.int64, device=dev),
            'start': torch.zeros(B, dtype=torch.int64, device=dev),
            'length': torch.zeros(B, dtype=torch.int64, device=dev),
            'outcome': torch.zeros(B, n_players, device=dev),
            'inv_total': torch.zeros(B, device=dev),
        }
        self.burn_in = int(trainer.args.get('burn_in_steps', 0))
        if self.burn_in:
            self.idx['lead'] = torch.zeros(B, dtype=torch.int64, device=dev)
        # initial hidden is always zeros: allocate once, reuse every step
        self.hidden0 = trainer.wrapped_model.init_hidden([B, n_players])
        if self.hidden0 is not None:
            from .util import map_r
            self.hidden0 = map_r(self.hidden0, lambda h: h.to(dev))
        self.graph = None
        if dev.type == 'cuda':
            for group in trainer.optimizer.param_groups:
                group['capturable'] = True
            trainer.model.train()
            stream = torch.cuda.Stream()
            stream.wait_stream(torch.cuda.current_stream())
            try:
                with torch.cuda.stream(stream):
                    for _ in range(warmup_iters):
                        self._fill()
                        self._run()
                torch.cuda.current_stream().wait_stream(stream)
                self.graph = torch.cuda.CUDAGraph()
                self._fill()
                with CAPTURE_LOCK, torch.cuda.graph(
                    self.graph, capture_error_mode='thread_local'):
                    self.losses, self.dcnt = self._run()
            except Exception as e:  # noqa: BLE001 - run eager if capture fails
                import sys
                print('recurrent train-step capture failed, running eager: %r'
                      % (e,), file=sys.stderr)
                self.graph = None

    def _fill(self):
        self.replay.publish()
        out = self.replay.sample_indices(self.batch_size)
        keys = ('pos0', 'start', 'length', 'outcome', 'inv_total') + \
            (('lead',) if self.burn_in else ())
        self._last_host_idx = dict(zip(keys, out))
        for key, arr in self._last_host_idx.items():
            # blocking H2D: `arr` is a temporary pageable numpy array; an
            # async copy that outlives it feeds garbage SAMPLE INDICES to
            # the captured ring gather (intermittent batch corruption)
            self.idx[key].copy_(torch.from_numpy(arr))

    def _run(self):
        tr = self.trainer
        batch = self.replay.gather_batch(
            self.idx['pos0'], self.idx['start'], self.idx['length'],
            self.idx['outcome'], self.idx['inv_total'],
            lead=self.idx.get('lead'))
        if tr.use_amp:
            with torch.autocast('cuda', dtype=torch.bfloat16):
                losses, dcnt = self._compute_loss(batch, tr.wrapped_model,
                                                  self.hidden0, tr.args)
        else:
            losses, dcnt = self._compute_loss(batch, tr.wrapped_model,
                                              self.hidden0, tr.args)
        tr.optimizer.zero_grad(set_to_none=False)
        losses['total'].backward()
        if tr.grad_guard:
            apply_grad_guard(tr.params, tr.guard_fires)
        tr.reducer.allreduce_()
        nn.utils.clip_grad_norm_(tr.params, 4.0)
        tr.optimizer.step()
        return losses, dcnt

    def step(self):
        """Sample indices on the host, copy them in, replay (or run)."""
        self._fill()
        self.trainer.steps += 1
        if self.graph is not None:
            self.graph.replay()
            return self.losses, self.dcnt
        return self._run()
