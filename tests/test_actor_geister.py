"""Batched Geister actor pool: valid reference-format episodes from the
multi-process recurrent path, trainable through the RNN learner."""

import numpy as np
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


def _pump_episodes(pool, env, n=3):
    model = env.net()
    model.eval()
    pool.attach(model, torch.device('cpu'))
    for _ in range(4000):
        pool.step_once()
        if pool.episodes_done >= n:
            break
    eps = pool.harvest()
    assert len(eps) >= n
    return eps


def _train_on(eps, env, args):
    buf = EpisodeBuffer(args)
    buf.extend(eps)
    trainer = Trainer(args, env.net(), device=torch.device('cpu'))
    batch = make_batch([buf.select_episode() for _ in range(4)], args)
    losses, dcnt = trainer.train_step(batch)
    assert torch.isfinite(losses['total'])
    assert dcnt > 0


def test_geister_pool_generates_and_trains_columnar():
    """Default path: vec engine + columnar turn-based episodes."""
    args = _args()
    pool = GeisterMultiProcPool(args, n_games=8, seed=3, workers=2)
    try:
        env = GeisterEnv()
        eps = _pump_episodes(pool, env)
        for ep in eps[:2]:
            assert ep['columnar'] and ep['turn_based']
            assert ep['steps'] >= 3
            assert set(ep['outcome'].keys()) == {0, 1}
            S = ep['steps']
            assert ep['board'].shape == (S, 7, 6, 6)
            assert ep['action'][0] >= 144         # layout turn comes first
            assert ep['turn'][0] == 0 and ep['turn'][1] == 1
            assert (ep['reward'] == np.float32(-0.01)).all()
            assert ep['return'].shape == (S, 2)
        _train_on(eps, env, args)
    finally:
        pool.shutdown()


def test_geister_pool_generates_and_trains_moment_dicts(monkeypatch):
    """Reference moment-dict format (HANDYRL_GEISTER_COLUMNAR=0)."""
    monkeypatch.setenv('HANDYRL_GEISTER_COLUMNAR', '0')
    args = _args()
    pool = GeisterMultiProcPool(args, n_games=8, seed=3, workers=2)
    try:
        env = GeisterEnv()
        eps = _pump_episodes(pool, env)
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
        _train_on(eps, env, args)
    finally:
        pool.shutdown()


def test_geister_pool_python_env_fallback(monkeypatch):
    """HANDYRL_GEISTER_VEC=0: the original python-Environment worker
    shards still generate trainable reference-format episodes."""
    monkeypatch.setenv('HANDYRL_GEISTER_VEC', '0')
    args = _args()
    pool = GeisterMultiProcPool(args, n_games=4, seed=6, workers=1)
    try:
        env = GeisterEnv()
        eps = _pump_episodes(pool, env, n=2)
        moments = unpack_moments({'moment': eps[0]['moment'], 'base': 0},
                                 0, eps[0]['steps'])
        assert len(moments) == eps[0]['steps']
        _train_on(eps, env, args)
    finally:
        pool.shutdown()


def test_geister_multiproc_pool_traj_mode_cpu():
    """Traj-mode pool end to end on CPU: workers ship fin metadata only,
    the engine records into GeisterTrajRecorder rings eagerly, and
    commit_traj fills the TurnDeviceReplay ring."""
    import torch
    from handyrl_amd.actor_geister import GeisterMultiProcPool
    from handyrl_amd.replay import TurnDeviceReplay
    from handyrl_amd.envs.geister import Environment

    args = {'turn_based_training': True, 'observation': False,
            'gamma': 0.8, 'forward_steps': 8, 'burn_in_steps': 0,
            'compress_steps': 4, 'batch_size': 4, 'minimum_episodes': 2,
            'maximum_episodes': 200, 'lambda': 0.7,
            'policy_target': 'UPGO', 'value_target': 'TD',
            'compress_episodes': False}
    pool = GeisterMultiProcPool(args, n_games=8, seed=11, workers=2,
                                traj_mode=True)
    dev = torch.device('cpu')
    replay = TurnDeviceReplay(args, dev, bytes_budget=32 << 20)
    try:
        torch.manual_seed(0)
        model = Environment().net()
        model.eval()
        pool.attach(model, dev, replay=replay)
        for _ in range(3000):
            pool.step_once()
            if pool.episodes_done >= 4:
                break
        assert pool.episodes_done >= 4
        assert len(replay) >= 4
        stubs = pool.harvest()
        assert stubs and all(s.get('committed') for s in stubs)
        # the committed rows sample and gather
        import numpy as np
        idx = replay.sample_indices(4)
        batch = replay.gather_batch(*[torch.from_numpy(np.asarray(a))
                                      for a in idx])
        assert torch.isfinite(batch['selected_prob']).all()
        assert batch['observation']['board'].shape[0] == 4
    finally:
        pool.shutdown()
