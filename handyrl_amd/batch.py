"""Replay storage, window sampling and training-batch assembly.

Batch semantics parity with reference train.py:33-124 (make_batch) and
train.py:270-318 (Batcher): (B, T, P, ...) tensors with burn-in/tail
padding (prob->1, action_mask->1e32, value-tail->outcome, progress->1),
recency-biased episode sampling, and forward-step windows block-aligned to
the compression block size.

Redesigned vs the reference: the episode store is an explicitly locked
ring buffer (the reference shares a bare deque across three threads,
reference train.py:323/472/293-301), and episode 'moment' blocks may be
either bz2-compressed pickles (remote workers) or plain lists (local GPU
actors skip compression entirely).
"""

import bz2
import pickle
import random
import threading
import time
from collections import deque

import numpy as np

from .connection import MultiProcessJobExecutor
from .model import to_torch
from .util import map_r, bimap_r, rotate


def unpack_moments(episode, start, end):
    """Decompress and slice an episode's moment blocks to [start, end)."""
    moments = []
    for block in episode['moment']:
        if isinstance(block, (bytes, bytearray)):
            moments.extend(pickle.loads(bz2.decompress(block)))
        else:
            moments.extend(block)
    base = episode['base']
    return moments[start - base:end - base]


def pack_moments(moments, compress_steps, compress=True):
    """Chunk moments into blocks of ``compress_steps`` (bz2-pickled or raw)."""
    blocks = []
    for i in range(0, len(moments), compress_steps):
        chunk = moments[i:i + compress_steps]
        blocks.append(bz2.compress(pickle.dumps(chunk)) if compress else chunk)
    return blocks


def _episode_fields_columnar(ep, args):
    """Per-episode batch fields from a COLUMNAR episode window.

    Columnar episodes (produced by the GPU actor pool) carry plain arrays
    (T,4,...) instead of per-step moment dicts: obs/alive/action/prob/value.
    Semantics mirror the dict path: a dead seat contributes zero obs/value,
    behavior prob 1, action 0 and a fully-illegal action mask.
    """
    alive = ep['alive']                          # (T, seats) bool
    n_seats = alive.shape[1]
    all_players = list(range(n_seats))
    if not args['turn_based_training']:
        players = [random.choice(all_players)]
    elif args['observation']:
        players = all_players
    else:
        raise NotImplementedError(
            'columnar episodes support solo or full-observation training')
    pl = np.array(players)
    T, NP = alive.shape[0], len(players)
    A = int(ep['n_actions'])

    aliveP = alive[:, pl]                                        # (T, P)
    if ep.get('canonical_obs'):
        # obs is (T, 17, 7, 11) canonical; seat views = CHMAP channel gather
        from .envs.vec_geese import CHMAP
        canon = ep['obs'].reshape(T, 17, 77)
        obs = canon[:, CHMAP[pl]].reshape(T, NP, 17, 7, 11)
    else:
        obs = ep['obs'][:, pl]
    obs = obs * aliveP[..., None, None, None].astype(obs.dtype)
    prob = np.where(aliveP, ep['prob'][:, pl], 1.0)[..., None].astype(np.float64)
    act = (ep['action'][:, pl] * aliveP).astype(np.int64)[..., None]
    amask = np.full((T, NP, A), 1e32, dtype=np.float32)
    amask[aliveP] = 0.0

    v = (ep['value'][:, pl] * aliveP)[..., None].astype(np.float32)
    rew = np.zeros((T, NP, 1), dtype=np.float32)
    ret = np.zeros((T, NP, 1), dtype=np.float32)
    oc = np.array([ep['outcome'][p] for p in players],
                  dtype=np.float32).reshape(1, NP, -1)
    emask = np.ones((T, 1, 1), dtype=np.float32)
    tmask = aliveP[..., None].astype(np.float32)
    omask = tmask.copy()
    progress = np.arange(ep['start'], ep['end'],
                         dtype=np.float32)[..., np.newaxis] / ep['total']
    obs_zeros = np.zeros_like(obs[0, 0])
    return (obs, obs_zeros, prob, v, act, oc, rew, ret, emask, tmask, omask,
            amask, progress)


def _episode_fields_columnar_turn(ep, args):
    """Per-episode batch fields from a COLUMNAR TURN-BASED episode window
    (observation=False configs, e.g. Geister): one trained seat per step
    (the mover), value/reward/return rows for every player.

    Shapes and dtypes are bit-identical to the moment-dict path (see
    tests/test_columnar_turn.py): obs/prob/action/action_mask carry the
    mover axis (T, 1, ...), value-side fields carry the player axis
    (T, P, ...) — reference train.py:57-110 make_batch semantics for the
    turn_based_training & not observation seat selection
    (train.py:65-68)."""
    assert args['turn_based_training'] and not args['observation'], \
        'turn-based columnar episodes need turn_based_training w/o observation'
    T = ep['turn'].shape[0]
    P = int(ep.get('n_players', 2))
    turn = ep['turn'].astype(np.int64)
    tr = np.arange(T)

    obs = {'scalar': ep['scalar'].astype(np.float32)[:, None, :],
           'board': ep['board'].astype(np.float32)[:, None]}
    prob = ep['prob'].astype(np.float64)[:, None, None]
    act = ep['action'].astype(np.int64)[:, None, None]
    if ep['mask'].dtype == np.bool_:          # legality bools -> additive mask
        amask = np.where(ep['mask'], np.float32(0), np.float32(1e32))[:, None]
    else:
        amask = ep['mask'].astype(np.float32)[:, None]
    v = np.zeros((T, P, 1), dtype=np.float32)
    v[tr, turn, 0] = ep['value']
    rew = ep['reward'].astype(np.float32).reshape(T, P, 1)
    ret = ep['return'].astype(np.float32).reshape(T, P, 1)
    oc = np.array([ep['outcome'][p] for p in range(P)],
                  dtype=np.float32).reshape(1, P, -1)
    emask = np.ones((T, 1, 1), dtype=np.float32)
    tmask = np.zeros((T, P, 1), dtype=np.float32)
    tmask[tr, turn, 0] = 1.0
    omask = tmask.copy()
    progress = np.arange(ep['start'], ep['end'],
                         dtype=np.float32)[..., np.newaxis] / ep['total']
    obs_zeros = {'scalar': np.zeros(ep['scalar'].shape[-1], dtype=np.float32),
                 'board': np.zeros(ep['board'].shape[2:], dtype=np.float32)}
    return (obs, obs_zeros, prob, v, act, oc, rew, ret, emask, tmask, omask,
            amask, progress)


def make_batch(episodes, args):
    """Assemble a (B, T, P, ...) training batch from sampled episode windows."""
    obss, datum = [], []

    def pick(a, b):
        return a if a is not None else b

    for ep in episodes:
        if ep.get('columnar'):
            fields_fn = _episode_fields_columnar_turn if ep.get('turn_based') \
                else _episode_fields_columnar
            (obs, obs_zeros, prob, v, act, oc, rew, ret, emask, tmask, omask,
             amask, progress) = fields_fn(ep, args)
            T = emask.shape[0]
            batch_steps = args['burn_in_steps'] + args['forward_steps']
            if T < batch_steps:
                pad_b = args['burn_in_steps'] - (ep['train_start'] - ep['start'])
                pad_a = batch_steps - T - pad_b
                pad3 = lambda a, val=0: np.pad(
                    a, [(pad_b, pad_a)] + [(0, 0)] * (a.ndim - 1),
                    'constant', constant_values=val)
                obs = map_r(obs, pad3)      # plain array or {scalar, board}
                prob = pad3(prob, val=1)
                v = np.concatenate(
                    [np.pad(v, [(pad_b, 0), (0, 0), (0, 0)], 'constant'),
                     np.tile(oc, [pad_a, 1, 1])])
                act = pad3(act)
                rew = pad3(rew)
                ret = pad3(ret)
                emask = pad3(emask)
                tmask = pad3(tmask)
                omask = pad3(omask)
                amask = pad3(amask, val=1e32)
                progress = np.pad(progress, [(pad_b, pad_a), (0, 0)],
                                  'constant', constant_values=1)
            obss.append(obs)
            datum.append((prob, v, act, oc, rew, ret, emask, tmask, omask,
                          amask, progress))
            continue

        moments = unpack_moments(ep, ep['start'], ep['end'])
        players = list(moments[0]['observation'].keys())
        if not args['turn_based_training']:     # solo training on one seat
            players = [random.choice(players)]

        turn0 = moments[0]['turn'][0]
        obs_zeros = map_r(moments[0]['observation'][turn0], lambda o: np.zeros_like(o))
        amask_zeros = np.zeros_like(moments[0]['action_mask'][turn0])

        # per-step seat list: the turn player only (turn-based, no observer
        # training) or every trained seat
        if args['turn_based_training'] and not args['observation']:
            seats_per_step = [[m['turn'][0]] for m in moments]
        else:
            seats_per_step = [players for _ in moments]

        obs = [[pick(m['observation'][s], obs_zeros) for s in seats]
               for m, seats in zip(moments, seats_per_step)]
        prob = np.array([[[pick(m['selected_prob'][s], 1.0)] for s in seats]
                         for m, seats in zip(moments, seats_per_step)])
        act = np.array([[pick(m['action'][s], 0) for s in seats]
                        for m, seats in zip(moments, seats_per_step)],
                       dtype=np.int64)[..., np.newaxis]
        amask = np.array([[pick(m['action_mask'][s], amask_zeros + 1e32) for s in seats]
                          for m, seats in zip(moments, seats_per_step)])

        # (T, P, ...) nests -> (..., T, P) leaf arrays
        obs = rotate(rotate(obs))
        obs = bimap_r(obs_zeros, obs, lambda _, o: np.array(o))

        T, NP = len(moments), len(players)
        v = np.array([[pick(m['value'][p], [0]) for p in players] for m in moments],
                     dtype=np.float32).reshape(T, NP, -1)
        rew = np.array([[pick(m['reward'][p], 0) for p in players] for m in moments],
                       dtype=np.float32).reshape(T, NP, -1)
        ret = np.array([[pick(m['return'][p], 0) for p in players] for m in moments],
                       dtype=np.float32).reshape(T, NP, -1)
        oc = np.array([ep['outcome'][p] for p in players],
                      dtype=np.float32).reshape(1, NP, -1)

        emask = np.ones((T, 1, 1), dtype=np.float32)
        tmask = np.array([[[m['selected_prob'][p] is not None] for p in players]
                          for m in moments], dtype=np.float32)
        omask = np.array([[[m['observation'][p] is not None] for p in players]
                          for m in moments], dtype=np.float32)
        progress = np.arange(ep['start'], ep['end'], dtype=np.float32)[..., np.newaxis] / ep['total']

        # short windows: pad with burn-in prefix zeros and a bootstrap tail
        batch_steps = args['burn_in_steps'] + args['forward_steps']
        if T < batch_steps:
            pad_b = args['burn_in_steps'] - (ep['train_start'] - ep['start'])
            pad_a = batch_steps - T - pad_b
            pad3 = lambda a, cb=0, ca=0, val=0: np.pad(
                a, [(pad_b, pad_a)] + [(0, 0)] * (a.ndim - 1), 'constant', constant_values=val)
            obs = map_r(obs, lambda o: np.pad(
                o, [(pad_b, pad_a)] + [(0, 0)] * (o.ndim - 1), 'constant'))
            prob = pad3(prob, val=1)
            v = np.concatenate(
                [np.pad(v, [(pad_b, 0), (0, 0), (0, 0)], 'constant'),
                 np.tile(oc, [pad_a, 1, 1])])
            act = pad3(act)
            rew = pad3(rew)
            ret = pad3(ret)
            emask = pad3(emask)
            tmask = pad3(tmask)
            omask = pad3(omask)
            amask = pad3(amask, val=1e32)
            progress = np.pad(progress, [(pad_b, pad_a), (0, 0)], 'constant', constant_values=1)

        obss.append(obs)
        datum.append((prob, v, act, oc, rew, ret, emask, tmask, omask, amask, progress))

    obs = to_torch(bimap_r(obs_zeros, rotate(obss), lambda _, o: np.array(o)))
    prob, v, act, oc, rew, ret, emask, tmask, omask, amask, progress = \
        [to_torch(np.array(val)) for val in zip(*datum)]

    return {
        'observation': obs,
        'selected_prob': prob,
        'value': v,
        'action': act, 'outcome': oc,
        'reward': rew, 'return': ret,
        'episode_mask': emask,
        'turn_mask': tmask, 'observation_mask': omask,
        'action_mask': amask,
        'progress': progress,
    }


class EpisodeBuffer:
    """Thread-safe bounded episode store with recency-biased sampling.

    Sampling: draw an index uniformly, accept with rate
    ``1 - (N-1-i)/N`` (newer episodes are kept more often), then cut a
    random forward-steps window, block-aligned to the compression blocks
    (reference train.py:291-315 semantics, made explicit and lock-guarded).
    """

    def __init__(self, args):
        self.args = args
        self.episodes = deque()
        self.lock = threading.Lock()
        self.total_added = 0

    def __len__(self):
        with self.lock:
            return len(self.episodes)

    def extend(self, episodes):
        with self.lock:
            self.episodes.extend(episodes)
            self.total_added += len(episodes)

    def trim(self, maximum):
        with self.lock:
            while len(self.episodes) > maximum:
                self.episodes.popleft()

    def select_episode(self):
        args = self.args
        while True:
            with self.lock:
                ep_count = min(len(self.episodes), args['maximum_episodes'])
                if ep_count > 0:
                    ep_idx = random.randrange(ep_count)
                    accept_rate = 1 - (ep_count - 1 - ep_idx) / ep_count
                    if random.random() < accept_rate:
                        ep = self.episodes[ep_idx]
                        break
                    continue
            time.sleep(0.01)
        turn_candidates = 1 + max(0, ep['steps'] - args['forward_steps'])
        train_st = random.randrange(turn_candidates)
        st = max(0, train_st - args['burn_in_steps'])
        ed = min(train_st + args['forward_steps'], ep['steps'])
        if ep.get('columnar'):
            # columnar episodes: the window is an array view, no block math
            if ep.get('turn_based'):
                step_keys = ('scalar', 'board', 'mask', 'turn', 'action',
                             'prob', 'value', 'reward', 'return')
            else:
                step_keys = ('obs', 'alive', 'action', 'prob', 'value')
            out = {
                'args': ep['args'], 'outcome': ep['outcome'], 'columnar': True,
                'turn_based': ep.get('turn_based', False),
                'canonical_obs': ep.get('canonical_obs', False),
                'n_actions': ep['n_actions'],
                'n_players': ep.get('n_players', 4),
                'start': st, 'end': ed, 'train_start': train_st,
                'total': ep['steps'],
            }
            for k in step_keys:
                out[k] = ep[k][st:ed]
            return out
        st_block = st // args['compress_steps']
        ed_block = (ed - 1) // args['compress_steps'] + 1
        return {
            'args': ep['args'], 'outcome': ep['outcome'],
            'moment': ep['moment'][st_block:ed_block],
            'base': st_block * args['compress_steps'],
            'start': st, 'end': ed, 'train_start': train_st, 'total': ep['steps'],
        }


class Batcher:
    """Parallel batch builders fed by the episode buffer."""

    def __init__(self, args, buffer):
        self.args = args
        self.buffer = buffer
        self.executor = MultiProcessJobExecutor(
            self._worker, self._selector(), self.args['num_batchers'])

    def _selector(self):
        while True:
            yield [self.buffer.select_episode() for _ in range(self.args['batch_size'])]

    def _worker(self, conn, bid):
        import sys
        print('started batcher %d' % bid, file=sys.stderr)
        while True:
            episodes = conn.recv()
            conn.send(make_batch(episodes, self.args))

    def run(self):
        self.executor.start()

    def batch(self):
        return self.executor.recv()
