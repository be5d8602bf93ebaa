"""DeviceReplay parity: the on-device sample-gather must reproduce the
columnar make_batch path for identical picks (run on CPU here; the same
torch ops run inside the training hipGraph on an MI355X)."""

import random

import numpy as np
import torch

from handyrl_amd.actor import GeeseActorPool
from handyrl_amd.batch import make_batch, EpisodeBuffer
from handyrl_amd.models.geese_net import GeeseNet
from handyrl_amd.replay import DeviceReplay


def _args(**over):
    args = {
        'turn_based_training': False, 'observation': False, 'gamma': 0.8,
        'forward_steps': 8, 'burn_in_steps': 0, 'compress_steps': 4,
        'entropy_regularization': 0.1, 'entropy_regularization_decay': 0.1,
        'batch_size': 6, 'minimum_episodes': 2, 'maximum_episodes': 100,
        'num_batchers': 1, 'lambda': 0.7, 'policy_target': 'VTRACE',
        'value_target': 'VTRACE', 'seed': 0, 'bf16': False,
        'compress_episodes': False,
    }
    args.update(over)
    return args


def _episodes(n=10, seed=0):
    args = _args()
    model = GeeseNet(layers=1)
    model.eval()
    pool = GeeseActorPool(model, args, n_games=6, device=torch.device('cpu'),
                          seed=seed)
    while pool.episodes_done < n:
        pool.step_once()
    return pool.harvest()[:n]


def test_device_replay_matches_make_batch():
    args = _args()
    episodes = _episodes(12)
    replay = DeviceReplay(args, torch.device('cpu'), bytes_budget=64 << 20)
    replay.extend(episodes)
    assert len(replay) == 12

    B = args['batch_size']
    random.seed(11)
    pos0, start, length, seat, outcome, inv_total = replay.sample_indices(B)

    dev = torch.device('cpu')
    batch_dev = replay.gather_batch(
        torch.from_numpy(pos0).to(dev), torch.from_numpy(start).to(dev),
        torch.from_numpy(length).to(dev), torch.from_numpy(seat).to(dev),
        torch.from_numpy(outcome).to(dev), torch.from_numpy(inv_total).to(dev))

    # rebuild the same picks through the reference columnar path
    table = list(replay.table)
    sels = []
    for b in range(B):
        # find the episode containing pos0
        ep_i = max(i for i, (p0, st, oc) in enumerate(table) if p0 <= pos0[b])
        p0, steps, oc = table[ep_i]
        ep = episodes[ep_i]
        st = int(start[b])
        ed = st + int(length[b])
        sels.append({
            'args': ep['args'], 'outcome': ep['outcome'], 'columnar': True,
            'canonical_obs': ep.get('canonical_obs', False), 'n_actions': 4,
            'obs': ep['obs'][st:ed], 'alive': ep['alive'][st:ed],
            'action': ep['action'][st:ed], 'prob': ep['prob'][st:ed],
            'value': ep['value'][st:ed],
            'start': st, 'end': ed, 'train_start': st, 'total': ep['steps'],
        })

    # make_batch picks a random solo seat; force the same seats
    class _FixedSeat:
        def __init__(self, seats):
            self.seats = list(seats)

        def __call__(self, players):
            return self.seats.pop(0)

    import handyrl_amd.batch as batch_mod
    orig_choice = batch_mod.random.choice
    batch_mod.random.choice = _FixedSeat(seat.tolist())
    try:
        batch_ref = make_batch(sels, args)
    finally:
        batch_mod.random.choice = orig_choice

    for key in batch_ref:
        td, tr = batch_dev[key], batch_ref[key]
        assert tuple(td.shape) == tuple(tr.shape), (key, td.shape, tr.shape)
        torch.testing.assert_close(td.double(), tr.double(), rtol=1e-5,
                                   atol=1e-5, msg=lambda m: '%s: %s' % (key, m))


def test_device_replay_ring_eviction():
    args = _args(maximum_episodes=1000)
    episodes = _episodes(12, seed=3)
    # tiny ring: forces overwrites
    replay = DeviceReplay(args, torch.device('cpu'), bytes_budget=1)
    assert replay.ring_T == 1024
    total_steps = 0
    for ep in episodes:
        replay.extend([ep])
        total_steps += ep['steps']
    # every retained episode's rows must still be valid (not overwritten)
    for p0, steps, oc in replay.table:
        assert p0 + steps <= replay.head
        assert p0 >= replay.head - replay.ring_T
    # sampling still works
    random.seed(0)
    out = replay.sample_indices(4)
    assert out[0].shape == (4,)
