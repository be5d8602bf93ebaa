"""TurnDeviceReplay parity: the on-device turn-based sample-gather must
reproduce make_batch's columnar turn-based path for identical picks (run
on CPU here; the same torch ops are hipGraph-capturable on an MI355X)."""

import random

import numpy as np
import torch

from handyrl_amd.actor_geister import GeisterActorPool
from handyrl_amd.batch import make_batch
from handyrl_amd.envs.geister import Environment as GeisterEnv
from handyrl_amd.replay import TurnDeviceReplay


def _args(**over):
    args = {
        'turn_based_training': True, 'observation': False, 'gamma': 0.8,
        'forward_steps': 8, 'burn_in_steps': 0, 'compress_steps': 4,
        'entropy_regularization': 0.1, 'entropy_regularization_decay': 0.1,
        'batch_size': 6, 'minimum_episodes': 2, 'maximum_episodes': 100,
        'num_batchers': 1, 'lambda': 0.7, 'policy_target': 'UPGO',
        'value_target': 'TD', 'seed': 0, 'bf16': False,
        'compress_episodes': False,
    }
    args.update(over)
    return args


def _episodes(n=10, seed=0):
    args = _args()
    model = GeisterEnv().net()
    model.eval()
    pool = GeisterActorPool(model, args, n_games=6,
                            device=torch.device('cpu'), seed=seed)
    while pool.episodes_done < n:
        pool.step_once()
    return pool.harvest()[:n]


def test_turn_device_replay_matches_make_batch():
    args = _args()
    episodes = _episodes(10)
    replay = TurnDeviceReplay(args, torch.device('cpu'), bytes_budget=64 << 20)
    replay.extend(episodes)
    assert len(replay) == 10

    B = args['batch_size']
    random.seed(23)
    pos0, start, length, outcome, inv_total = replay.sample_indices(B)

    dev = torch.device('cpu')
    batch_dev = replay.gather_batch(
        torch.from_numpy(pos0).to(dev), torch.from_numpy(start).to(dev),
        torch.from_numpy(length).to(dev),
        torch.from_numpy(outcome).to(dev), torch.from_numpy(inv_total).to(dev))

    # rebuild the same picks through the reference columnar path
    table = list(replay.table)
    sels = []
    for b in range(B):
        ep_i = max(i for i, (p0, _st, _oc) in enumerate(table)
                   if p0 <= pos0[b])
        ep = episodes[ep_i]
        st = int(start[b])
        ed = st + int(length[b])
        sel = {'args': ep['args'], 'outcome': ep['outcome'], 'columnar': True,
               'turn_based': True, 'n_actions': ep['n_actions'],
               'n_players': 2, 'start': st, 'end': ed, 'train_start': st,
               'total': ep['steps']}
        for k in ('scalar', 'board', 'mask', 'turn', 'action', 'prob',
                  'value', 'reward', 'return'):
            sel[k] = ep[k][st:ed]
        sels.append(sel)
    batch_ref = make_batch(sels, args)

    def check(key, td, tr):
        assert tuple(td.shape) == tuple(tr.shape), (key, td.shape, tr.shape)
        torch.testing.assert_close(td.double(), tr.double(), rtol=1e-5,
                                   atol=1e-5,
                                   msg=lambda m: '%s: %s' % (key, m))

    for key in batch_ref:
        if isinstance(batch_ref[key], dict):
            for kk in batch_ref[key]:
                check(key + '.' + kk, batch_dev[key][kk], batch_ref[key][kk])
        else:
            check(key, batch_dev[key], batch_ref[key])


def test_turn_device_replay_trains():
    from handyrl_amd.train import Trainer, compute_loss
    args = _args(batch_size=4)
    episodes = _episodes(8, seed=4)
    replay = TurnDeviceReplay(args, torch.device('cpu'), bytes_budget=64 << 20)
    replay.extend(episodes)
    random.seed(1)
    idx = replay.sample_indices(4)
    batch = replay.gather_batch(*[torch.from_numpy(a) for a in idx])
    trainer = Trainer(args, GeisterEnv().net(), device=torch.device('cpu'))
    hidden = trainer.wrapped_model.init_hidden([4, 2])
    losses, dcnt = compute_loss(batch, trainer.wrapped_model, hidden, args)
    assert torch.isfinite(losses['total'])
    losses['total'].backward()
    grads = [p.grad for p in trainer.model.parameters() if p.grad is not None]
    assert grads and all(torch.isfinite(g).all() for g in grads)


def test_recurrent_replay_train_step_runs():
    """GraphedRecurrentTrainStep end to end on CPU (eager path; capture is
    attempted only on GPU): sample -> device gather -> RNN forward ->
    loss -> backward -> Adam, losses finite over several steps."""
    from handyrl_amd.hipgraph import GraphedRecurrentTrainStep
    from handyrl_amd.train import Trainer
    args = _args(batch_size=3)
    episodes = _episodes(8, seed=9)
    replay = TurnDeviceReplay(args, torch.device('cpu'), bytes_budget=64 << 20)
    replay.extend(episodes)
    trainer = Trainer(args, GeisterEnv().net(), device=torch.device('cpu'))
    step = GraphedRecurrentTrainStep(trainer, replay, args['batch_size'])
    random.seed(2)
    for _ in range(3):
        losses, dcnt = step.step()
        assert torch.isfinite(losses['total'])
        assert float(dcnt) > 0
    assert trainer.steps == 3


def test_turn_device_replay_burnin_matches_make_batch():
    """burn_in_steps > 0: the device gather spans burn_in+forward_steps
    with per-sample leading pads, matching make_batch's prefix padding."""
    args = _args(forward_steps=6, burn_in_steps=4)
    episodes = _episodes(8, seed=13)
    replay = TurnDeviceReplay(args, torch.device('cpu'), bytes_budget=64 << 20)
    replay.extend(episodes)

    B = 6
    random.seed(31)
    pos0, start, length, outcome, inv_total, lead = replay.sample_indices(B)
    dev = torch.device('cpu')
    batch_dev = replay.gather_batch(
        torch.from_numpy(pos0).to(dev), torch.from_numpy(start).to(dev),
        torch.from_numpy(length).to(dev), torch.from_numpy(outcome).to(dev),
        torch.from_numpy(inv_total).to(dev), torch.from_numpy(lead).to(dev))

    table = list(replay.table)
    sels = []
    for b in range(B):
        ep_i = max(i for i, (p0, _st, _oc) in enumerate(table)
                   if p0 <= pos0[b])
        ep = episodes[ep_i]
        st = int(start[b])
        ed = st + int(length[b])
        train_st = st + args['burn_in_steps'] - int(lead[b])
        sel = {'args': ep['args'], 'outcome': ep['outcome'], 'columnar': True,
               'turn_based': True, 'n_actions': ep['n_actions'],
               'n_players': 2, 'start': st, 'end': ed,
               'train_start': train_st, 'total': ep['steps']}
        for k in ('scalar', 'board', 'mask', 'turn', 'action', 'prob',
                  'value', 'reward', 'return'):
            sel[k] = ep[k][st:ed]
        sels.append(sel)
    batch_ref = make_batch(sels, args)

    for key in batch_ref:
        if isinstance(batch_ref[key], dict):
            for kk in batch_ref[key]:
                td, tr = batch_dev[key][kk], batch_ref[key][kk]
                assert tuple(td.shape) == tuple(tr.shape), (key, kk, td.shape,
                                                            tr.shape)
                torch.testing.assert_close(td.double(), tr.double(),
                                           rtol=1e-5, atol=1e-5)
        else:
            td, tr = batch_dev[key], batch_ref[key]
            assert tuple(td.shape) == tuple(tr.shape), (key, td.shape, tr.shape)
            torch.testing.assert_close(td.double(), tr.double(), rtol=1e-5,
                                       atol=1e-5,
                                       msg=lambda m, k=key: '%s: %s' % (k, m))


def test_recurrent_replay_train_step_with_burnin():
    from handyrl_amd.hipgraph import GraphedRecurrentTrainStep
    from handyrl_amd.train import Trainer
    args = _args(batch_size=3, forward_steps=6, burn_in_steps=3)
    episodes = _episodes(8, seed=21)
    replay = TurnDeviceReplay(args, torch.device('cpu'), bytes_budget=64 << 20)
    replay.extend(episodes)
    trainer = Trainer(args, GeisterEnv().net(), device=torch.device('cpu'))
    step = GraphedRecurrentTrainStep(trainer, replay, args['batch_size'])
    random.seed(5)
    for _ in range(2):
        losses, dcnt = step.step()
        assert torch.isfinite(losses['total'])
        assert float(dcnt) > 0
