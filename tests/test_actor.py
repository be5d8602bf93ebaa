"""GPU actor pool tests (CPU execution here; the same code runs batched
bf16 inference on an MI355X)."""

import numpy as np
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
    for _ in range(220):
        pool.step_once()
        if pool.episodes_done >= 4:
            break
    episodes = pool.harvest()
    assert len(episodes) >= 4
    for ep in episodes:
        assert ep['steps'] >= 1
        assert set(ep['outcome'].keys()) == {0, 1, 2, 3}
        assert abs(sum(ep['outcome'].values())) < 1e-6    # pairwise zero sum
        moments = [m for block in ep['moment'] for m in block]
        assert len(moments) == ep['steps']
        m0 = moments[0]
        assert sorted(m0['turn']) == m0['turn']
        for p in m0['turn']:
            assert m0['observation'][p].shape == (17, 7, 11)
            assert m0['observation'][p].dtype == np.uint8
            assert 0 <= m0['action'][p] < 4
            assert 0 < m0['selected_prob'][p] <= 1


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
