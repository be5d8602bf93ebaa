"""Batched Geister actor pool: valid reference-format episodes from the
multi-process recurrent path, trainable through the RNN learner."""

import torch

from handyrl_amd.actor_geister import GeisterMultiProcPool
from handyrl_amd.batch import make_batch, EpisodeBuffer, unpack_moments  # noqa: F401
from handyrl_amd.envs.geister import Environment as GeisterEnv
from handyrl_amd.train import Trainer


def _args(**over):
    args = {
        'turn_based_training': True, 'observation': False, 'gamma': 0.8,
        'forward_steps': 4, 'burn_in_steps': 0, 'compress_steps': 4,
        'entropy_regularization': 0.1, 'entropy_regularization_decay': 0.1,
        'batch_size': 4, 'minimum_episodes': 2, 'maximum_episodes': 50,
        'num_batchers': 1, 'lambda': 0.7, 'policy_target': 'UPGO',
        'value_target': 'TD', 'seed': 0, 'bf16': False,
        'compress_episodes': False,
    }
    args.update(over)
    return args


def test_geister_pool_generates_and_trains():
    args = _args()
    pool = GeisterMultiProcPool(args, n_games=8, seed=3, workers=2)
    try:
        env = GeisterEnv()
        model = env.net()
        model.eval()
        pool.attach(model, torch.device('cpu'))
        for _ in range(4000):
            pool.step_once()
            if pool.episodes_done >= 3:
                break
        eps = pool.harvest()
        assert len(eps) >= 3
        for ep in eps[:2]:
            assert ep['steps'] >= 3
            assert set(ep['outcome'].keys()) == {0, 1}
            moments = unpack_moments({'moment': ep['moment'], 'base': 0},
                                     0, ep['steps'])
            assert len(moments) == ep['steps']
            m0 = moments[0]
            p = m0['turn'][0]
            assert m0['observation'][p]['board'].shape == (7, 6, 6)
            assert m0['action'][p] >= 144         # layout turn comes first
            assert m0['reward'][0] == -0.01
            # returns backfilled for both seats
            assert moments[-1]['return'][0] is not None

        buf = EpisodeBuffer(args)
        buf.extend(eps)
        trainer = Trainer(args, env.net(), device=torch.device('cpu'))
        batch = make_batch([buf.select_episode() for _ in range(4)], args)
        losses, dcnt = trainer.train_step(batch)
        assert torch.isfinite(losses['total'])
        assert dcnt > 0
    finally:
        pool.shutdown()
