"""GPU actor pool tests (CPU execution here; the same code runs batched
bf16 inference + hipGraphs on an MI355X)."""

import random

import numpy as np
import pytest
import torch

from handyrl_amd.actor import GeeseActorPool
from handyrl_amd.batch import make_batch, EpisodeBuffer
from handyrl_amd.models.geese_net import GeeseNet
from handyrl_amd.train import Trainer


def _args(**over):
    args = {
        'turn_based_training': False, 'observation': False, 'gamma': 0.8,
        'forward_steps': 8, 'burn_in_steps': 0, 'compress_steps': 4,
        'entropy_regularization': 0.1, 'entropy_regularization_decay': 0.1,
        'batch_size': 4, 'minimum_episodes': 2, 'maximum_episodes': 100,
        'num_batchers': 1, 'lambda': 0.7, 'policy_target': 'VTRACE',
        'value_target': 'VTRACE', 'seed': 0, 'bf16': False,
        'compress_episodes': False,
    }
    args.update(over)
    return args


def test_actor_pool_generates_valid_episodes():
    args = _args()
    model = GeeseNet(layers=2)
    model.eval()
    pool = GeeseActorPool(model, args, n_games=8, device=torch.device('cpu'), seed=1)
    for _ in range(250):
        pool.step_once()
        if pool.episodes_done >= 4:
            break
    episodes = pool.harvest()
    assert len(episodes) >= 4
    for ep in episodes:
        S = ep['steps']
        assert S >= 1
        assert ep['columnar']
        assert set(ep['outcome'].keys()) == {0, 1, 2, 3}
        assert abs(sum(ep['outcome'].values())) < 1e-6    # pairwise zero sum
        assert ep['canonical_obs']
        assert ep['obs'].shape == (S, 17, 7, 11) and ep['obs'].dtype == np.uint8
        assert ep['alive'].shape == (S, 4)
        assert ep['alive'][0].all()                        # all alive at start
        live = ep['alive']
        assert ((ep['action'] >= 0) & (ep['action'] < 4))[live].all()
        assert (ep['prob'][live] > 0).all() and (ep['prob'][live] <= 1).all()
        # dead seats never revive
        for p in range(4):
            col = live[:, p].astype(int)
            assert (np.diff(col) <= 0).all()


def _dict_episode_from_columnar(ep):
    """Reference-format (moment dict) episode with identical content."""
    from handyrl_amd.envs.vec_geese import CHMAP
    moments = []
    S = ep['steps']
    # canonical obs -> per-seat views for the dict format
    obs_seat = ep['obs'].reshape(S, 17, 77)[:, CHMAP].reshape(S, 4, 17, 7, 11)
    for t in range(S):
        keys = ('observation', 'selected_prob', 'action_mask', 'action',
                'value', 'reward', 'return')
        moment = {k: {p: None for p in range(4)} for k in keys}
        turn = [p for p in range(4) if ep['alive'][t, p]]
        for p in turn:
            moment['observation'][p] = obs_seat[t, p]
            moment['selected_prob'][p] = float(ep['prob'][t, p])
            moment['action_mask'][p] = np.zeros(4, dtype=np.float32)
            moment['action'][p] = int(ep['action'][t, p])
            moment['value'][p] = [float(ep['value'][t, p])]
        moment['turn'] = turn
        moments.append(moment)
    return {
        'args': ep['args'], 'steps': S, 'outcome': ep['outcome'],
        'moment': [moments[i:i + 4] for i in range(0, S, 4)],
    }


def test_columnar_batch_matches_dict_batch():
    """make_batch must produce identical tensors from the columnar fast
    path and the reference moment-dict path."""
    args = _args(forward_steps=6)
    model = GeeseNet(layers=1)
    model.eval()
    pool = GeeseActorPool(model, args, n_games=4, device=torch.device('cpu'), seed=3)
    while pool.episodes_done < 3:
        pool.step_once()
    cols = pool.harvest()[:3]

    buf_c = EpisodeBuffer(args)
    buf_c.extend(cols)
    buf_d = EpisodeBuffer(args)
    buf_d.extend([_dict_episode_from_columnar(ep) for ep in cols])

    random.seed(42)
    sel_c = [buf_c.select_episode() for _ in range(4)]
    random.seed(42)
    sel_d = [buf_d.select_episode() for _ in range(4)]
    for c, d in zip(sel_c, sel_d):
        assert (c['start'], c['end'], c['train_start']) == \
            (d['start'], d['end'], d['train_start'])

    random.seed(7)
    batch_c = make_batch(sel_c, args)
    random.seed(7)
    batch_d = make_batch(sel_d, args)
    for key in batch_d:
        tc, td = batch_c[key], batch_d[key]
        assert tc.shape == td.shape, key
        torch.testing.assert_close(tc.double(), td.double(), rtol=1e-6,
                                   atol=1e-6, msg=lambda m: '%s: %s' % (key, m))


def test_actor_episodes_train():
    """Episodes from the GPU-actor path feed the standard learner."""
    args = _args()
    model = GeeseNet(layers=2)
    pool = GeeseActorPool(model, args, n_games=8, device=torch.device('cpu'), seed=2)
    while pool.episodes_done < 4:
        pool.step_once()
    buf = EpisodeBuffer(args)
    buf.extend(pool.harvest())
    trainer = Trainer(args, GeeseNet(layers=2), device=torch.device('cpu'))
    batch = make_batch([buf.select_episode() for _ in range(args['batch_size'])], args)
    assert batch['observation'].dtype == torch.uint8
    losses, dcnt = trainer.train_step(batch)
    assert dcnt > 0
    assert torch.isfinite(losses['total'])


def test_multiproc_pool_cpu():
    """Env-worker processes + shared-memory transport, CPU inference."""
    from handyrl_amd.actor import MultiProcGeesePool
    args = _args()
    mpool = MultiProcGeesePool(args, n_games=12, seed=5, workers=2)
    try:
        model = GeeseNet(layers=1)
        model.eval()
        mpool.attach(model, torch.device('cpu'))
        for _ in range(900):
            mpool.step_once()
            if mpool.episodes_done >= 6:
                break
        eps = mpool.harvest()
        assert len(eps) >= 6
        for ep in eps[:3]:
            assert ep['columnar'] and ep['steps'] >= 1
            assert abs(sum(ep['outcome'].values())) < 1e-6
        assert mpool.frames > 0
        # episodes train
        buf = EpisodeBuffer(args)
        buf.extend(eps)
        trainer = Trainer(args, GeeseNet(layers=1), device=torch.device('cpu'))
        batch = make_batch([buf.select_episode() for _ in range(4)], args)
        losses, dcnt = trainer.train_step(batch)
        assert torch.isfinite(losses['total'])
    finally:
        mpool.shutdown()
