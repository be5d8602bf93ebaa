"""Learner: batching, loss computation, optimization and orchestration.

Semantics parity with reference train.py (forward_prediction :127-186,
compose_losses :189-215, compute_loss :218-267, Trainer :321-400,
Learner :403-645) with the MI355X re-architecture:

* device learner with bf16 autocast compute, fp32 loss math and fp32
  master weights (Adam);
* data-parallel replicas are separate PROCESSES, one per GPU, synchronized
  by a single fused RCCL all-reduce of gradients + the data-count scalar
  over xGMI (handyrl_amd/dist.py) — replacing nn.DataParallel
  (reference train.py:339-340);
* the off-policy target scans run as one fused HIP kernel on GPU
  (handyrl_amd/losses.py -> handyrl_amd/ops);
* the episode store is an explicitly locked buffer (handyrl_amd/batch.py)
  instead of a shared bare deque.
"""

import copy
import os
import queue
import random
import threading
import time
import warnings

import torch
import torch.nn as nn
import torch.nn.functional as F
import torch.optim as optim

try:
    import psutil
except ImportError:       # pragma: no cover - psutil is in the image
    psutil = None

from .environment import prepare_env, make_env
from .util import map_r, bimap_r, trimap_r
from .model import to_gpu, ModelWrapper
from .losses import compute_target
from .batch import EpisodeBuffer, Batcher
from .worker import WorkerCluster, WorkerServer
from . import dist as hdist


def forward_prediction(model, hidden, batch, args):
    """Run the network over a (B, T, P) batch; FF nets flatten to one batch
    axis, RNNs loop T with observation-masked hidden carry and burn-in under
    no_grad."""
    observations = batch['observation']
    # GPU actors store 0/1 observations as uint8 (4x H2D traffic saved);
    # cast to compute dtype at the consumer
    observations = map_r(
        observations,
        lambda o: o.float() if not torch.is_floating_point(o) else o)
    batch_shape = batch['action'].size()[:3]    # (B, T, P or 1)

    if hidden is None:
        obs = map_r(observations, lambda o: o.flatten(0, 2))
        outputs = model(obs, None)
        outputs = map_r(outputs, lambda o: o.unflatten(0, batch_shape))
    else:
        outputs = {}
        for t in range(batch_shape[1]):
            obs = map_r(observations, lambda o: o[:, t].flatten(0, 1))
            omask_t = batch['observation_mask'][:, t]
            omask = map_r(hidden, lambda h: omask_t.view(*h.size()[:2], *([1] * (h.dim() - 2))))
            hidden_masked = bimap_r(hidden, omask, lambda h, m: h * m)
            if args['turn_based_training'] and not args['observation']:
                hidden_in = map_r(hidden_masked, lambda h: h.sum(1))
            else:
                hidden_in = map_r(hidden_masked, lambda h: h.flatten(0, 1))
            if t < args['burn_in_steps']:
                model.eval()
                with torch.no_grad():
                    outputs_t = model(obs, hidden_in)
            else:
                if not model.training:
                    model.train()
                outputs_t = model(obs, hidden_in)
            outputs_t = map_r(outputs_t, lambda o: o.unflatten(0, (batch_shape[0], batch_shape[2])))
            next_hidden = None
            for k, o in outputs_t.items():
                if k == 'hidden':
                    next_hidden = o
                else:
                    outputs.setdefault(k, []).append(o)
            hidden = trimap_r(hidden, next_hidden, omask, lambda h, nh, m: h * (1 - m) + nh * m)
        outputs = {k: torch.stack(o, dim=1) for k, o in outputs.items() if o[0] is not None}

    for k, o in outputs.items():
        if k == 'policy':
            o = o.mul(batch['turn_mask'])
            if o.size(2) > 1 and batch_shape[2] == 1:
                o = o.sum(2, keepdim=True)        # gather the turn seat
            outputs[k] = o - batch['action_mask']
        else:
            outputs[k] = o.mul(batch['observation_mask'])
    return outputs


def compose_losses(outputs, log_selected_policies, total_advantages, targets, batch, args):
    """Summed (not averaged) loss terms + the data count that scales lr."""
    tmasks = batch['turn_mask']
    omasks = batch['observation_mask']

    losses = {}
    dcnt = tmasks.sum()          # tensor: stays device-side (hipGraph-safe)

    losses['p'] = (-log_selected_policies * total_advantages).mul(tmasks).sum()
    if 'value' in outputs:
        losses['v'] = ((outputs['value'] - targets['value']) ** 2).mul(omasks).sum() / 2
    if 'return' in outputs:
        losses['r'] = F.smooth_l1_loss(outputs['return'], targets['return'],
                                       reduction='none').mul(omasks).sum()

    # entropy = -(p * log p) over the (finitely) masked logits; written in
    # primitives because Categorical's arg validation host-syncs, which
    # breaks hipGraph capture
    logp = F.log_softmax(outputs['policy'], dim=-1)
    entropy = -(logp.exp() * logp).sum(-1).mul(tmasks.sum(-1))
    losses['ent'] = entropy.sum()

    base_loss = losses['p'] + losses.get('v', 0) + losses.get('r', 0)
    decay = 1 - batch['progress'] * (1 - args['entropy_regularization_decay'])
    entropy_loss = entropy.mul(decay).sum() * -args['entropy_regularization']
    losses['total'] = base_loss + entropy_loss
    return losses, dcnt


class _FusedLossHead(torch.autograd.Function):
    """Fused loss pipeline for the flagship family (feed-forward solo,
    value head only): the ~40 eager ops around the target scans
    (importance ratios, value/outcome splice, pg/value/entropy sums and
    their backward) run as three HIP kernels + the fused scan
    (ops/src/ext.hip loss_head_*; reference semantics train.py:229-267,
    189-215).  forward returns accum (5,) = [p, v, ent, ent*decay, dcnt];
    the caller assembles total = p + v - ent_reg * ent_decay_sum."""

    @staticmethod
    def forward(ctx, policy, value, action, mu, emask, tmask, omask,
                outcome, ret, progress, shapes, args):
        from . import ops as _ops
        ext = _ops.require()
        B, T = shapes
        log_sel, rho, v_spl = ext.loss_head_pre(
            policy, action, mu, emask, value.detach(), outcome)
        view = lambda t: t.view(B, T, 1, 1)
        oc4 = outcome.view(B, T, 1, 1)
        targets_v, adv_v = compute_target(
            args['value_target'], view(v_spl), oc4, None,
            args['lambda'], 1, view(rho), view(rho), view(omask))
        if args['policy_target'] != args['value_target']:
            _, adv_v = compute_target(
                args['policy_target'], view(v_spl), oc4, None,
                args['lambda'], 1, view(rho), view(rho), view(omask))
        targets_v = targets_v.reshape(-1).contiguous()
        # 'return'-head-less family: target == advantage == batch return
        ta = (rho * (adv_v.reshape(-1) + ret)).contiguous()
        accum = ext.loss_head_fwd(policy, log_sel, ta, tmask, omask,
                                  value.detach(), targets_v, progress,
                                  args['entropy_regularization_decay'])
        ctx.save_for_backward(policy, action, ta, tmask, omask, emask,
                              value.detach(), targets_v, progress)
        ctx.ent_decay = args['entropy_regularization_decay']
        return accum

    @staticmethod
    def backward(ctx, gaccum):
        from . import ops as _ops
        ext = _ops.require()
        (policy, action, ta, tmask, omask, emask, value, targets_v,
         progress) = ctx.saved_tensors
        dpolicy, dvalue = ext.loss_head_bwd(
            policy, action, ta, tmask, omask, emask, value, targets_v,
            progress, gaccum.contiguous(), ctx.ent_decay)
        return (dpolicy, dvalue) + (None,) * 10


def _fused_loss_ok(outputs, batch, args):
    return (outputs['policy'].is_cuda
            and os.environ.get('HANDYRL_FUSED_LOSS', '1') == '1'
            and 'value' in outputs and 'return' not in outputs
            and outputs['policy'].dim() == 4
            and outputs['policy'].size(2) == 1       # solo seat axis
            and not args['turn_based_training'])


def _compute_loss_fused(outputs, batch, args):
    """compute_loss tail on the fused loss-head kernels (flagship
    family); numerics parity vs the eager tail is pinned by
    tests/test_gpu.py::test_fused_loss_head_matches_eager."""
    policy = outputs['policy'].float()
    value = outputs['value'].float()
    B, T = policy.size(0), policy.size(1)
    A = policy.size(-1)
    flat = lambda t: t.reshape(-1).contiguous().float()
    oc = (batch['outcome'] * torch.ones_like(batch['value'])).reshape(-1)
    accum = _FusedLossHead.apply(
        policy.reshape(-1, A).contiguous(), value.reshape(-1),
        batch['action'].reshape(-1).contiguous().long(),
        flat(batch['selected_prob']), flat(batch['episode_mask']),
        flat(batch['turn_mask']), flat(batch['observation_mask']),
        oc.contiguous().float(), flat(batch['return']),
        flat(batch['progress']), (B, T), args)
    losses = {'p': accum[0], 'v': accum[1], 'ent': accum[2]}
    losses['total'] = accum[0] + accum[1] \
        - accum[3] * args['entropy_regularization']
    return losses, accum[4]


def compute_loss(batch, model, hidden, args):
    outputs = forward_prediction(model, hidden, batch, args)
    if args['burn_in_steps'] > 0:
        batch = map_r(batch, lambda v: v[:, args['burn_in_steps']:] if v.size(1) > 1 else v)
        outputs = map_r(outputs, lambda v: v[:, args['burn_in_steps']:])

    if _fused_loss_ok(outputs, batch, args):
        from . import ops as _ops
        if _ops.available():
            return _compute_loss_fused(outputs, batch, args)

    # loss math in fp32 regardless of autocast compute dtype
    outputs = {k: o.float() for k, o in outputs.items()}

    actions = batch['action']
    emasks = batch['episode_mask']
    omasks = batch['observation_mask']
    value_target_masks, return_target_masks = omasks, omasks

    clip_rho_threshold, clip_c_threshold = 1.0, 1.0

    log_selected_b_policies = torch.log(torch.clamp(batch['selected_prob'], 1e-16, 1)) * emasks
    log_selected_t_policies = F.log_softmax(outputs['policy'], dim=-1) \
        .gather(-1, actions) * emasks

    # importance weights (behavior vs current policy), clipped
    log_rhos = log_selected_t_policies.detach() - log_selected_b_policies
    rhos = torch.exp(log_rhos)
    clipped_rhos = torch.clamp(rhos, 0, clip_rho_threshold)
    cs = torch.clamp(rhos, 0, clip_c_threshold)
    outputs_nograd = {k: o.detach() for k, o in outputs.items()}

    if 'value' in outputs_nograd:
        values_nograd = outputs_nograd['value']
        if args['turn_based_training'] and values_nograd.size(2) == 2:
            # two-player zero-sum: symmetrize with the opponent's negated view
            values_opp = -torch.flip(values_nograd, dims=[2])
            omasks_opp = torch.flip(omasks, dims=[2])
            values_nograd = (values_nograd * omasks + values_opp * omasks_opp) \
                / (omasks + omasks_opp + 1e-8)
            value_target_masks = torch.clamp(omasks + omasks_opp, 0, 1)
        # splice the terminal outcome in past the episode end
        outputs_nograd['value'] = values_nograd * emasks + batch['outcome'] * (1 - emasks)

    targets, advantages = {}, {}
    value_args = (outputs_nograd.get('value', None), batch['outcome'], None,
                  args['lambda'], 1, clipped_rhos, cs, value_target_masks)
    return_args = (outputs_nograd.get('return', None), batch['return'], batch['reward'],
                   args['lambda'], args['gamma'], clipped_rhos, cs, return_target_masks)

    targets['value'], advantages['value'] = compute_target(args['value_target'], *value_args)
    targets['return'], advantages['return'] = compute_target(args['value_target'], *return_args)
    if args['policy_target'] != args['value_target']:
        _, advantages['value'] = compute_target(args['policy_target'], *value_args)
        _, advantages['return'] = compute_target(args['policy_target'], *return_args)

    total_advantages = clipped_rhos * sum(advantages.values())
    return compose_losses(outputs, log_selected_t_policies, total_advantages,
                          targets, batch, args)


def apply_grad_guard(params, counter=None):
    """Finite-guard gradients in place, counting activations.

    Pure tensor ops (hipGraph-capturable): when any gradient holds a
    non-finite value, `counter` (a device int64 scalar) is incremented by
    one for the step, so silently-absorbed spike steps are observable
    (the Trainer reports the count each epoch)."""
    flag = None
    for p in params:
        g = p.grad
        if g is None:
            continue
        if counter is not None:
            nf = (~torch.isfinite(g)).sum()
            flag = nf if flag is None else flag + nf
        torch.nan_to_num_(g, nan=0.0, posinf=1e6, neginf=-1e6)
    if counter is not None and flag is not None:
        counter.add_((flag > 0).to(counter.dtype))


class Trainer:
    """SGD loop over batches from the episode buffer.

    Multi-GPU: each DP rank owns one Trainer on its own GPU; gradients and
    the per-epoch data count are SUM-all-reduced so the dynamic lr
    (default_lr * data_cnt_ema, decayed by steps) sees the global batch.
    """

    def __init__(self, args, model, device=None, episodes=None, batcher=None):
        self.args = args
        self.device = device if device is not None else (
            torch.device('cuda', hdist.env_local_rank())
            if torch.cuda.is_available() else torch.device('cpu'))
        self.use_amp = bool(args.get('bf16', True)) and self.device.type == 'cuda'

        self.episodes = episodes if episodes is not None else EpisodeBuffer(args)
        self.template_model = copy.deepcopy(model).cpu()
        self.model = model.to(self.device)
        if self.device.type == 'cuda' and getattr(model, 'prefers_channels_last', False):
            self.model = self.model.to(memory_format=torch.channels_last)
        self.params = list(self.model.parameters())
        self.reducer = hdist.GradReducer(self.params)
        # Finite-guard the gradients before clipping: a single spike step
        # (observed under policy-saturation trajectories: loss still finite,
        # grads inf -> clip_grad_norm scales inf*0 -> NaN weights, after
        # which the whole self-play loop is poisoned — diagnosis in
        # BASELINE.md "learning sanity") becomes a survivable clipped
        # update instead of permanent NaN.  HANDYRL_GRAD_GUARD=0 disables.
        self.grad_guard = os.environ.get('HANDYRL_GRAD_GUARD', '1') == '1'
        # device-resident activation counter (readable without breaking
        # graph capture; reported per epoch in train())
        self.guard_fires = torch.zeros((), dtype=torch.int64,
                                       device=self.device)
        self._guard_reported = 0

        self.default_lr = 3e-8
        self.data_cnt_ema = args['batch_size'] * args['forward_steps']
        lr = self.default_lr * self.data_cnt_ema
        # On GPU the lr lives as a device tensor so a CAPTURED Adam step
        # still sees each epoch's dynamic-lr update (reference
        # train.py:382-384 semantics) without re-capturing the graph:
        # the replayed kernels read the tensor, we fill_() it in place.
        if self.device.type == 'cuda' and \
                os.environ.get('HANDYRL_TENSOR_LR', '1') == '1':
            lr = torch.tensor(lr, dtype=torch.float32, device=self.device)
        self.optimizer = optim.Adam(self.params, lr=lr, weight_decay=1e-5) \
            if len(self.params) > 0 else None
        self.steps = 0
        # batch-builder processes fork HERE (construction time, main thread,
        # before worker/server threads start): forking later from a threaded
        # process risks a child deadlock on inherited locks.  Callers that
        # need the fork even earlier (before any HIP context, e.g. bench.py)
        # pass a pre-built Batcher in.
        # replay: 'device' — an HBM-resident replay ring replaces BOTH the
        # host episode buffer and the multiprocess batch builders; train()
        # then runs the (captured) sample-gather train step.  Supported for
        # feed-forward solo configs (DeviceReplay) and turn-based
        # observation=False recurrent configs (TurnDeviceReplay).
        self.device_replay = None
        self._replay_step = None
        replay_mode = args.get('replay')
        if (replay_mode is None and self.device.type == 'cuda'
                and args.get('worker', {}).get('type') == 'gpu'
                and args['turn_based_training']
                and not args['observation']
                and not args.get('burn_in_steps', 0)):
            # measured default: the HBM replay ring + captured recurrent
            # step runs Geister GPU-actor configs 1.75x the batcher path
            # (34.7k vs 19.8k frames/s at 512 actors);
            # set replay: 'host' to force the multiprocess batchers
            replay_mode = 'device'
            print("replay: 'device' (auto: GPU-actor turn-based config)")
        if replay_mode == 'device':
            from .replay import DeviceReplay, TurnDeviceReplay
            if args.get('burn_in_steps', 0) and \
                    not args['turn_based_training']:
                raise ValueError(
                    "replay: 'device' supports burn_in_steps only on the "
                    "turn-based path")
            budget = int(args.get('replay_bytes',
                                  (4 << 30) if self.device.type == 'cuda'
                                  else (64 << 20)))
            ingest = self.device.type == 'cuda'
            if args['turn_based_training']:
                if args['observation']:
                    raise ValueError(
                        "replay: 'device' supports solo or turn-based "
                        "observation=False configs")
                self.device_replay = TurnDeviceReplay(
                    args, self.device, bytes_budget=budget,
                    ingest_thread=ingest)
            else:
                self.device_replay = DeviceReplay(
                    args, self.device, bytes_budget=budget,
                    ingest_thread=ingest)
            self.episodes = self.device_replay
            self.batcher = False
        else:
            self.batcher = batcher if batcher is not None \
                else Batcher(args, self.episodes)
        self.update_flag = threading.Event()
        self.update_queue = queue.Queue(maxsize=1)
        self.wrapped_model = ModelWrapper(self.model)

    def update(self):
        self.update_flag.set()
        model, steps = self.update_queue.get()
        return model, steps

    def snapshot(self):
        """CPU eval-mode copy of the current weights (no device churn)."""
        snap = copy.deepcopy(self.template_model)
        snap.load_state_dict(
            {k: v.detach().cpu() for k, v in self.model.state_dict().items()})
        snap.eval()
        return snap

    def enable_cuda_graph(self, example_batch_cpu):
        """Capture the whole train step into a hipGraph (FF models, fixed
        batch shapes).  Returns True on success; on failure the eager path
        stays active."""
        if self.device.type != 'cuda':
            return False
        try:
            from .hipgraph import GraphedTrainStep
            self.graphed_step = GraphedTrainStep(self, example_batch_cpu)
            return True
        except Exception as e:        # noqa: BLE001 - deliberate fallback
            import sys
            print('cuda-graph capture failed, using eager train step: %r' % (e,),
                  file=sys.stderr)
            self.graphed_step = None
            return False

    def train_step(self, batch):
        """One optimizer step on an already-built CPU batch dict."""
        graphed = getattr(self, 'graphed_step', None)
        if graphed is not None:
            losses_t, dcnt_t = graphed.step(batch)
            return losses_t, float(dcnt_t)
        batch_size = batch['value'].size(0)
        player_count = batch['value'].size(2)
        hidden = self.wrapped_model.init_hidden([batch_size, player_count])
        if self.device.type == 'cuda':
            batch = to_gpu(batch, non_blocking=True)
            if hidden is not None:
                hidden = map_r(hidden, lambda h: h.cuda(non_blocking=True))

        if self.use_amp:
            with torch.autocast('cuda', dtype=torch.bfloat16):
                losses, dcnt = compute_loss(batch, self.wrapped_model, hidden, self.args)
        else:
            losses, dcnt = compute_loss(batch, self.wrapped_model, hidden, self.args)

        self.optimizer.zero_grad(set_to_none=False)
        losses['total'].backward()
        if self.grad_guard:
            apply_grad_guard(self.params, self.guard_fires)
        self.reducer.allreduce_()                 # fused RCCL all-reduce (DP)
        nn.utils.clip_grad_norm_(self.params, 4.0)
        self.optimizer.step()
        self.steps += 1
        return losses, float(dcnt)

    def train(self):
        if self.optimizer is None:
            time.sleep(0.1)
            return self.snapshot()

        batch_cnt, data_cnt, loss_sum = 0, 0, {}
        self.model.train()

        while data_cnt == 0 or not self.update_flag.is_set():
            if self.device_replay is not None:
                losses, dcnt = self._device_replay_step()
                dcnt = float(dcnt)
            else:
                batch = self.batcher.batch()
                losses, dcnt = self.train_step(batch)
            batch_cnt += 1
            data_cnt += dcnt
            for k, l in losses.items():
                loss_sum[k] = loss_sum.get(k, 0.0) + l.item()

        print('loss = %s' % ' '.join(
            [k + ':' + '%.3f' % (l / data_cnt) for k, l in loss_sum.items()]))
        if self.grad_guard:
            fires = int(self.guard_fires.item())
            if fires > self._guard_reported:
                print('grad guard fired on %d step(s) this epoch '
                      '(%d total)' % (fires - self._guard_reported, fires))
                self._guard_reported = fires

        global_data_cnt = hdist.allreduce_scalar(data_cnt / (1e-2 + batch_cnt))
        self.data_cnt_ema = self.data_cnt_ema * 0.8 + global_data_cnt * 0.2
        new_lr = self.default_lr * self.data_cnt_ema / (1 + self.steps * 1e-5)
        for param_group in self.optimizer.param_groups:
            if torch.is_tensor(param_group['lr']):
                param_group['lr'].fill_(new_lr)
            else:
                param_group['lr'] = new_lr
        return self.snapshot()

    def _device_replay_step(self):
        """One step on the device-replay path (lazy build: the captured
        gather-train step needs a non-empty ring)."""
        if self._replay_step is None:
            from .hipgraph import (GraphedReplayTrainStep,
                                   GraphedRecurrentTrainStep)
            if self.args['turn_based_training']:
                self._replay_step = GraphedRecurrentTrainStep(
                    self, self.device_replay, self.args['batch_size'])
            else:
                self._replay_step = GraphedReplayTrainStep(
                    self, self.device_replay, self.args['batch_size'])
        return self._replay_step.step()

    def run(self):
        print('waiting training')
        while len(self.episodes) < self.args['minimum_episodes']:
            if self.device_replay is not None:
                # admit background-ingested blocks so the count advances
                self.device_replay.publish()
            time.sleep(1)
        if self.optimizer is not None:
            if self.batcher:
                self.batcher.run()
            print('started training')
        while True:
            model = self.train()
            self.update_flag.clear()
            self.update_queue.put((model, self.steps))


class ScoreBook:
    """(count, sum, sum-of-squares) accumulators keyed by (epoch, tag).

    Backs the win-rate and generation-stats reports; the stdout line
    grammar is a compatibility contract with the plot tools
    (scripts/win_rate_plot.py, stats_plot.py)."""

    def __init__(self):
        self._acc = {}

    def add(self, key, value, tag=None):
        n, s, s2 = self._acc.get((key, tag), (0, 0.0, 0.0))
        self._acc[(key, tag)] = (n + 1, s + value, s2 + value * value)

    def tags(self, key):
        return sorted(t for (k, t) in self._acc if k == key and t is not None)

    def has(self, key, tag=None):
        return (key, tag) in self._acc

    def total(self, key, tag=None):
        return self._acc.get((key, tag), (0, 0.0, 0.0))

    def print_win_rate(self, key, tag=None, label=''):
        n, r, _ = self.total(key, tag)
        mean = r / (n + 1e-6)
        name_tag = ' (%s)' % label if label != '' else ''
        print('win rate%s = %.3f (%.1f / %d)'
              % (name_tag, (mean + 1) / 2, (r + n) / 2, n))

    def print_generation(self, key):
        n, r, r2 = self.total(key)
        mean = r / (n + 1e-6)
        std = (r2 / (n + 1e-6) - mean ** 2) ** 0.5
        print('generation stats = %.3f +- %.3f' % (mean, std))


class Learner:
    """Central conductor: owns the model epoch, serves worker requests
    (job args / episodes / results / model pulls) and rolls training
    epochs.

    Protocol and report-line parity with the reference Learner
    (reference train.py:403-645): job dicts carry role/player/model_id,
    the eval share follows eval_rate with seat rotation, model pulls
    return pickled modules, and the epoch report prints the win-rate /
    generation-stats grammar the plot tools parse.  MI355X extensions:
    a GPU actor mode (self-play as batched inference on the learner GPU,
    CPU workers demoted to evaluation), optimizer-state sidecars, and
    reference-layout checkpoint export."""

    def __init__(self, args, net=None, remote=False):
        train_args = args['train_args']
        env_args = args['env_args']
        train_args['env'] = env_args
        args = train_args

        self.args = args
        random.seed(args['seed'])

        self.env = make_env(env_args)
        # never let the eval share starve at tiny update_episodes
        floor = (args['update_episodes'] ** 0.85) / args['update_episodes']
        self.eval_rate = max(args['eval_rate'], floor)
        self.shutdown_flag = False
        self.flags = set()

        self.model_epoch = self.args['restart_epoch']
        self.model = net if net is not None else self.env.net()
        if self.model_epoch > 0:
            sd = torch.load(self.model_path(self.model_epoch))
            if hasattr(self.model, 'load_reference_state_dict'):
                self.model.load_reference_state_dict(sd)
            else:
                self.model.load_state_dict(sd, strict=False)

        self.generation_stats = ScoreBook()
        self.eval_stats = ScoreBook()
        self.num_episodes = 0
        self.num_returned_episodes = 0
        self.num_results = 0

        # GPU actor pool (worker: {type: 'gpu'}): self-play generation runs
        # as batched inference on the learner's GPU instead of CPU worker
        # processes; any CPU workers then serve evaluation jobs only.
        # HungryGeese uses the multiprocess env-worker pool (the flagship
        # bench architecture: vectorized native env cores in child
        # processes + one shared inference engine) — it must FORK before
        # the Trainer creates a HIP context.
        self.gpu_actor = bool(args['worker'].get('type') == 'gpu')
        self._mpool = None
        env_name = str(env_args.get('env'))
        if self.gpu_actor and torch.cuda.is_available():
            n_envs = args['worker'].get('num_envs', 2048)
            procs = int(args['worker'].get('num_actor_procs', 8))
            if env_name == 'HungryGeese':
                from .actor import MultiProcGeesePool
                self._mpool = MultiProcGeesePool(
                    args, n_games=n_envs, seed=args['seed'] + 1,
                    workers=procs,
                    traj_mode=args.get('replay') == 'device')
            elif env_name == 'Geister':
                from .actor_geister import GeisterMultiProcPool
                # Geister ships episodes over the pipe into the device
                # replay (measured faster than device trajectory
                # recording for this env: 151k vs 146k at 2048 actors;
                # HANDYRL_GEISTER_TRAJ=1 opts into the traj rings)
                traj = os.environ.get('HANDYRL_GEISTER_TRAJ', '0') == '1'
                self._mpool = GeisterMultiProcPool(
                    args, n_games=n_envs, seed=args['seed'] + 1,
                    workers=procs, traj_mode=traj)

        self.worker = WorkerServer(args) if remote else WorkerCluster(args)
        self.trainer = Trainer(args, copy.deepcopy(self.model))
        # [amd] save_optimizer: restore the Adam state / step counters
        # saved alongside restart_epoch (exact resume; the reference
        # restarts the optimizer cold, train.py:420-423)
        if self.model_epoch > 0 and self.trainer.optimizer is not None:
            opt_path = self.optimizer_path(self.model_epoch)
            if self.args.get('save_optimizer') and os.path.exists(opt_path):
                st = torch.load(opt_path)
                self.trainer.optimizer.load_state_dict(st['optimizer'])
                self.trainer.steps = st['steps']
                self.trainer.data_cnt_ema = st['data_cnt_ema']
                print('restored optimizer state at epoch %d' % self.model_epoch)

        self.feed_lock = threading.Lock()

    # -- checkpoint files --------------------------------------------------

    def model_path(self, model_id):
        return os.path.join('models', str(model_id) + '.pth')

    def latest_model_path(self):
        return os.path.join('models', 'latest.pth')

    def optimizer_path(self, model_id):
        # [amd] extension sidecar; the .pth files stay plain state_dicts
        # (reference train.py:441-454 layout is a compatibility contract)
        return os.path.join('models', str(model_id) + '.opt.pth')

    def update_model(self, model, steps):
        print('updated model(%d)' % steps)
        self.model_epoch += 1
        self.model = model
        os.makedirs('models', exist_ok=True)
        # nets that differ structurally from their reference counterpart
        # (e.g. GeeseNet drops the redundant under-BN conv biases) export a
        # reference-layout state dict so saved .pth files load into the
        # reference net unchanged (the .pth layout is a compat contract)
        if hasattr(model, 'reference_state_dict'):
            sd = model.reference_state_dict()
        else:
            sd = model.state_dict()
        torch.save(sd, self.model_path(self.model_epoch))
        torch.save(sd, self.latest_model_path())
        if self.args.get('save_optimizer') and \
                self.trainer.optimizer is not None:
            torch.save({'optimizer': self.trainer.optimizer.state_dict(),
                        'steps': self.trainer.steps,
                        'data_cnt_ema': self.trainer.data_cnt_ema},
                       self.optimizer_path(self.model_epoch))

    # -- data intake -------------------------------------------------------

    def feed_episodes(self, episodes):
        live = [e for e in episodes if e is not None]
        with self.feed_lock:
            for episode in live:
                for p in episode['args']['player']:
                    self.generation_stats.add(self.model_epoch,
                                              episode['outcome'][p])
                self.num_returned_episodes += 1
                if self.gpu_actor:
                    self.num_episodes += 1
                if self.num_returned_episodes % 100 == 0:
                    print(self.num_returned_episodes, end=' ', flush=True)
            self.trainer.episodes.extend(live)
            self._ram_watchdog()

    def _ram_watchdog(self):
        """Shrink the episode buffer under host-RAM pressure (reference
        train.py:474-483 semantics); called under feed_lock."""
        mem_percent = psutil.virtual_memory().percent if psutil else 0
        limit = self.args['maximum_episodes']
        if mem_percent > 95:
            limit = int(len(self.trainer.episodes) * 95 / mem_percent)
            if 'memory_over' not in self.flags:
                warnings.warn('memory usage %.1f%% with buffer size %d'
                              % (mem_percent, len(self.trainer.episodes)))
                self.flags.add('memory_over')
        self.trainer.episodes.trim(limit)

    def feed_results(self, results):
        for result in results:
            if result is None:
                continue
            for p in result['args']['player']:
                res = result['result'][p]
                self.eval_stats.add(self.model_epoch, res)
                self.eval_stats.add(self.model_epoch, res,
                                    tag=result['opponent'])

    # -- GPU actor mode ----------------------------------------------------

    def _gpu_actor_loop(self):
        """Self-play generation on the learner GPU (worker type 'gpu'):
        a vectorized env pool stepped by batched (graphed) inference on the
        trainer's live model, feeding columnar episodes straight into the
        replay buffer — no per-env worker processes, no pickled models."""
        n_envs = self.args['worker'].get('num_envs', 256)
        env_name = str(self.args.get('env', {}).get('env', ''))
        if self._mpool is not None:
            pool = self._mpool
            pool.attach(self.trainer.model, self.trainer.device,
                        replay=self.trainer.device_replay)
            n_envs = self.args['worker'].get('num_envs', 2048)
        elif env_name == 'Geister':
            from .actor_geister import GeisterActorPool
            pool = GeisterActorPool(self.trainer.model, self.args,
                                    n_games=n_envs,
                                    device=self.trainer.device,
                                    seed=self.args['seed'] + 1)
        elif env_name == 'HungryGeese':
            from .actor import GeeseActorPool
            pool = GeeseActorPool(self.trainer.model, self.args,
                                  n_games=n_envs,
                                  device=self.trainer.device,
                                  seed=self.args['seed'] + 1)
        else:
            raise ValueError(
                "worker type 'gpu' supports HungryGeese and Geister "
                "(got env=%r); use CPU workers for other envs" % env_name)
        last_epoch = -1
        print('started gpu actor pool (%d envs)' % n_envs)
        while not self.shutdown_flag:
            if self.model_epoch != last_epoch:
                pool.refresh_weights()
                last_epoch = self.model_epoch
            was_training = self.trainer.model.training
            self.trainer.model.eval()
            from .hipgraph import CAPTURE_LOCK
            with CAPTURE_LOCK:
                # never replay while the trainer thread captures its graph
                for _ in range(8):
                    pool.step_once()
            if was_training:
                self.trainer.model.train()
            episodes = pool.harvest()
            if episodes:
                self.feed_episodes(episodes)
        print('finished gpu actor pool')

    # -- epoch rollover ----------------------------------------------------

    def update(self):
        print()
        print('epoch %d' % self.model_epoch)
        self._print_report()
        model, steps = self.trainer.update()
        if model is None:
            model = self.model
        self.update_model(model, steps)
        self.flags = set()

    def _print_report(self):
        ep = self.model_epoch
        if not self.eval_stats.has(ep):
            print('win rate = Nan (0)')
        else:
            tags = self.eval_stats.tags(ep)
            single_opponent = \
                len(self.args.get('eval', {}).get('opponent', [])) <= 1
            if single_opponent and len(tags) <= 1:
                self.eval_stats.print_win_rate(ep)
            else:
                self.eval_stats.print_win_rate(ep, label='total')
                for tag in tags:
                    self.eval_stats.print_win_rate(ep, tag=tag, label=tag)
        if not self.generation_stats.has(ep):
            print('generation stats = Nan (0)')
        else:
            self.generation_stats.print_generation(ep)

    # -- request serving ---------------------------------------------------

    def _next_job(self):
        """Assign one worker job (reference train.py:566-587 policy):
        evaluation until its share catches up with eval_rate — always,
        when GPU actors generate — otherwise generation; eval jobs rotate
        the evaluated seat and mark opponent seats with model_id -1."""
        players = self.env.players()
        job = {'model_id': {}}
        evaluate = self.gpu_actor or \
            self.num_results < self.eval_rate * self.num_episodes
        if evaluate:
            job['role'] = 'e'
            seat = players[self.num_results % len(players)]
            job['player'] = [seat]
            for p in players:
                job['model_id'][p] = self.model_epoch if p == seat else -1
            self.num_results += 1
        else:
            job['role'] = 'g'
            job['player'] = players
            for p in players:
                job['model_id'][p] = self.model_epoch
            self.num_episodes += 1
        return job

    def _pickled_model(self, model_id):
        model = self.model
        if model_id != self.model_epoch and model_id > 0:
            try:
                model = copy.deepcopy(self.model)
                model.load_state_dict(
                    torch.load(self.model_path(model_id)), strict=False)
            except Exception:
                pass   # fall back to the latest model
        import pickle
        return pickle.dumps(model)

    def _serve(self, req, payloads):
        if req == 'args':
            if self.shutdown_flag:
                return [None] * len(payloads)
            return [self._next_job() for _ in payloads]
        if req == 'episode':
            self.feed_episodes(payloads)
            return [None] * len(payloads)
        if req == 'result':
            self.feed_results(payloads)
            return [None] * len(payloads)
        if req == 'model':
            return [self._pickled_model(mid) for mid in payloads]
        return [None] * len(payloads)

    def _maybe_rollover(self, state):
        """Epoch rollover bookkeeping; returns True to leave the server
        loop (GPU-actor shutdown with no workers attached)."""
        if self.shutdown_flag:
            # GPU actors can buffer tens of thousands of episodes; without
            # this guard the backlog keeps re-firing rollovers past the
            # configured epoch count during the drain
            return self.gpu_actor and self.worker.connection_count() == 0
        if self.num_returned_episodes < state['next_update']:
            return False
        state['next_update'] += self.args['update_episodes']
        self.update()
        if 0 <= self.args['epochs'] <= self.model_epoch:
            self.shutdown_flag = True
            if self.gpu_actor and self.worker.connection_count() == 0:
                return True
        return False

    def server(self):
        print('started server')
        state = {'next_update': self.args['minimum_episodes'] +
                 self.args['update_episodes']}
        while self.worker.connection_count() > 0 or not self.shutdown_flag:
            # rollover is checked every iteration: with GPU actors the
            # buffer fills without any worker request traffic
            if self._maybe_rollover(state):
                break
            try:
                conn, (req, payload) = self.worker.recv(timeout=0.3)
            except queue.Empty:
                continue
            batched = isinstance(payload, list)
            replies = self._serve(req, payload if batched else [payload])
            self.worker.send(conn, replies if batched else replies[0])
        print('finished server')

    def run(self):
        threading.Thread(target=self.trainer.run, daemon=True).start()
        self._actor_thread = None
        if self.gpu_actor:
            self._actor_thread = threading.Thread(
                target=self._gpu_actor_loop, daemon=True)
            self._actor_thread.start()
        # a LOCAL cluster with zero workers has nothing to spawn; the remote
        # WorkerServer must always run (it accepts workers that join later)
        if isinstance(self.worker, WorkerServer) or \
                self.args['worker'].get('num_parallel', 0) > 0:
            self.worker.run()
        try:
            self.server()
        finally:
            self.shutdown_flag = True
            if self._actor_thread is not None:
                self._actor_thread.join(timeout=10)
            if self._mpool is not None:
                self._mpool.shutdown()    # quit the env workers cleanly


def train_main(args):
    prepare_env(args['env_args'])
    learner = Learner(args=args)
    learner.run()


def train_server_main(args):
    learner = Learner(args=args, remote=True)
    learner.run()
