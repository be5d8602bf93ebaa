"""Learner-core tests: single train steps (FF and RNN paths), loss
finiteness, and a full local --train integration run in a subprocess."""

import os
import random
import subprocess
import sys
import tempfile
import textwrap

import pytest
import torch

from handyrl_amd.batch import make_batch, EpisodeBuffer
from handyrl_amd.generation import Generator
from handyrl_amd.model import ModelWrapper
from handyrl_amd.train import Trainer, compute_loss
from handyrl_amd.envs import tictactoe, geister, hungry_geese

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _args(**over):
    args = {
        'turn_based_training': True,
        'observation': False,
        'gamma': 0.8,
        'forward_steps': 8,
        'burn_in_steps': 0,
        'compress_steps': 4,
        'entropy_regularization': 0.1,
        'entropy_regularization_decay': 0.1,
        'batch_size': 4,
        'minimum_episodes': 2,
        'maximum_episodes': 100,
        'num_batchers': 1,
        'lambda': 0.7,
        'policy_target': 'TD',
        'value_target': 'TD',
        'seed': 0,
        'bf16': False,
        'compress_episodes': True,
    }
    args.update(over)
    return args


def _episodes(env_mod, args, n=4):
    episodes = []
    env = env_mod.Environment()
    gen = Generator(env, args)
    models = {p: ModelWrapper(env.net()) for p in env.players()}
    job = {'player': env.players(), 'model_id': {p: 1 for p in env.players()}}
    for i in range(n):
        random.seed(i)
        ep = gen.generate(models, job)
        assert ep is not None
        episodes.append(ep)
    return episodes


def _one_batch(env_mod, args, n=3):
    buf = EpisodeBuffer(args)
    buf.extend(_episodes(env_mod, args, n))
    sel = [buf.select_episode() for _ in range(args['batch_size'])]
    return make_batch(sel, args)


@pytest.mark.parametrize('algo', ['TD', 'VTRACE', 'UPGO', 'MC'])
def test_train_step_tictactoe(algo):
    args = _args(policy_target=algo, value_target=algo)
    env = tictactoe.Environment()
    trainer = Trainer(args, env.net(), device=torch.device('cpu'))
    batch = _one_batch(tictactoe, args)
    before = [p.detach().clone() for p in trainer.params]
    losses, dcnt = trainer.train_step(batch)
    assert dcnt > 0
    for k, l in losses.items():
        assert torch.isfinite(l), (k, l)
    assert any(not torch.equal(b, p.detach()) for b, p in zip(before, trainer.params))


def test_train_step_geese_solo():
    args = _args(turn_based_training=False)
    env = hungry_geese.Environment()
    trainer = Trainer(args, env.net(), device=torch.device('cpu'))
    batch = _one_batch(hungry_geese, args, n=2)
    losses, dcnt = trainer.train_step(batch)
    assert dcnt > 0
    assert torch.isfinite(losses['total'])


def test_compute_loss_geister_rnn_burn_in():
    """RNN path: per-timestep loop, hidden masking, burn-in slicing."""
    args = _args(forward_steps=4, burn_in_steps=2, compress_steps=2,
                 policy_target='UPGO', value_target='TD')
    env = geister.Environment()
    model = env.net()
    wrapped = ModelWrapper(model)
    batch = _one_batch(geister, args, n=2)
    B, P = batch['value'].size(0), batch['value'].size(2)
    hidden = wrapped.init_hidden([B, P])
    losses, dcnt = compute_loss(batch, wrapped, hidden, args)
    assert torch.isfinite(losses['total'])
    losses['total'].backward()
    grads = [p.grad for p in model.parameters() if p.grad is not None]
    assert len(grads) > 0
    assert all(torch.isfinite(g).all() for g in grads)


def test_full_gpu_actor_mode_training_run():
    """--train with worker type 'gpu': generation via the vectorized actor
    pool in the learner process (CPU device here), CPU workers eval-only."""
    script = textwrap.dedent('''
        import sys
        sys.path.insert(0, %r)
        from handyrl_amd.train import train_main
        args = {
            'env_args': {'env': 'HungryGeese'},
            'train_args': {
                'turn_based_training': False, 'observation': False,
                'gamma': 0.8, 'forward_steps': 8, 'burn_in_steps': 0,
                'compress_steps': 4, 'entropy_regularization': 0.1,
                'entropy_regularization_decay': 0.1, 'update_episodes': 30,
                'batch_size': 4, 'minimum_episodes': 10, 'maximum_episodes': 300,
                'epochs': 1, 'num_batchers': 1, 'eval_rate': 0.1,
                'worker': {'type': 'gpu', 'num_parallel': 1, 'num_envs': 8},
                'lambda': 0.7, 'policy_target': 'VTRACE', 'value_target': 'VTRACE',
                'eval': {'opponent': ['random']}, 'seed': 0, 'restart_epoch': 0,
                'bf16': False,
            },
        }
        train_main(args)
        print('TRAIN_DONE')
    ''') % REPO
    with tempfile.TemporaryDirectory() as tmp:
        res = subprocess.run([sys.executable, '-c', script], cwd=tmp,
                             capture_output=True, text=True, timeout=300)
        out = res.stdout
        assert 'TRAIN_DONE' in out, (out[-3000:], res.stderr[-3000:])
        assert 'started gpu actor pool' in out
        assert 'updated model(' in out
        assert os.path.exists(os.path.join(tmp, 'models', '1.pth'))


def test_full_local_training_run():
    """main.py --train equivalent: learner + gather + workers, one epoch."""
    script = textwrap.dedent('''
        import sys
        sys.path.insert(0, %r)
        from handyrl_amd.train import train_main
        args = {
            'env_args': {'env': 'TicTacToe'},
            'train_args': {
                'turn_based_training': True, 'observation': False,
                'gamma': 0.8, 'forward_steps': 4, 'burn_in_steps': 0,
                'compress_steps': 4, 'entropy_regularization': 0.1,
                'entropy_regularization_decay': 0.1, 'update_episodes': 15,
                'batch_size': 4, 'minimum_episodes': 5, 'maximum_episodes': 200,
                'epochs': 1, 'num_batchers': 1, 'eval_rate': 0.1,
                'worker': {'num_parallel': 2}, 'lambda': 0.7,
                'policy_target': 'TD', 'value_target': 'TD',
                'eval': {'opponent': ['random']}, 'seed': 0, 'restart_epoch': 0,
                'bf16': False,
            },
        }
        train_main(args)
        print('TRAIN_DONE')
    ''') % REPO
    with tempfile.TemporaryDirectory() as tmp:
        res = subprocess.run([sys.executable, '-c', script], cwd=tmp,
                             capture_output=True, text=True, timeout=300)
        out = res.stdout
        assert 'TRAIN_DONE' in out, (out[-3000:], res.stderr[-3000:])
        assert 'updated model(' in out
        assert os.path.exists(os.path.join(tmp, 'models', '1.pth'))
        assert os.path.exists(os.path.join(tmp, 'models', 'latest.pth'))


def test_restart_epoch_resume():
    """Checkpoint/resume: restart_epoch=N loads models/N.pth and continues
    numbering at N+1 (reference train.py:420-423 semantics)."""
    base_args = {
        'env_args': {'env': 'TicTacToe'},
        'train_args': {
            'turn_based_training': True, 'observation': False,
            'gamma': 0.8, 'forward_steps': 4, 'burn_in_steps': 0,
            'compress_steps': 4, 'entropy_regularization': 0.1,
            'entropy_regularization_decay': 0.1, 'update_episodes': 12,
            'batch_size': 4, 'minimum_episodes': 5, 'maximum_episodes': 200,
            'epochs': 1, 'num_batchers': 1, 'eval_rate': 0.1,
            'worker': {'num_parallel': 2}, 'lambda': 0.7,
            'policy_target': 'TD', 'value_target': 'TD',
            'eval': {'opponent': ['random']}, 'seed': 0, 'restart_epoch': 0,
            'bf16': False, 'save_optimizer': True,
        },
    }
    script = textwrap.dedent('''
        import sys, json
        sys.path.insert(0, %r)
        from handyrl_amd.train import train_main
        args = json.loads(%r)
        args['train_args']['restart_epoch'] = int(sys.argv[1])
        args['train_args']['epochs'] = int(sys.argv[2])
        train_main(args)
        print('TRAIN_DONE')
    ''')
    import json as _json
    script = script % (REPO, _json.dumps(base_args))
    with tempfile.TemporaryDirectory() as tmp:
        r1 = subprocess.run([sys.executable, '-c', script, '0', '1'], cwd=tmp,
                            capture_output=True, text=True, timeout=240)
        assert 'TRAIN_DONE' in r1.stdout, (r1.stdout[-2000:], r1.stderr[-2000:])
        assert os.path.exists(os.path.join(tmp, 'models', '1.pth'))
        # [amd] save_optimizer sidecar written next to the checkpoint
        assert os.path.exists(os.path.join(tmp, 'models', '1.opt.pth'))
        r2 = subprocess.run([sys.executable, '-c', script, '1', '2'], cwd=tmp,
                            capture_output=True, text=True, timeout=240)
        assert 'TRAIN_DONE' in r2.stdout, (r2.stdout[-2000:], r2.stderr[-2000:])
        assert os.path.exists(os.path.join(tmp, 'models', '2.pth'))
        assert 'epoch 1' in r2.stdout       # resumed at the loaded epoch
        assert 'restored optimizer state at epoch 1' in r2.stdout


def test_full_gpu_actor_mode_training_run_geister():
    """--train with worker type 'gpu' on the RECURRENT Geister env: the
    in-process GeisterActorPool (vec rules engine + batched DRC inference
    + columnar turn-based episodes) generates inside the learner."""
    script = textwrap.dedent('''
        import sys
        sys.path.insert(0, %r)
        from handyrl_amd.train import train_main
        args = {
            'env_args': {'env': 'Geister'},
            'train_args': {
                'turn_based_training': True, 'observation': False,
                'gamma': 0.8, 'forward_steps': 8, 'burn_in_steps': 0,
                'compress_steps': 4, 'entropy_regularization': 0.1,
                'entropy_regularization_decay': 0.1, 'update_episodes': 20,
                'batch_size': 2, 'minimum_episodes': 4, 'maximum_episodes': 200,
                'epochs': 1, 'num_batchers': 1, 'eval_rate': 0.1,
                'worker': {'type': 'gpu', 'num_parallel': 1, 'num_envs': 8},
                'lambda': 0.7, 'policy_target': 'UPGO', 'value_target': 'TD',
                'eval': {'opponent': ['random']}, 'seed': 0, 'restart_epoch': 0,
                'bf16': False,
            },
        }
        train_main(args)
        print('TRAIN_DONE')
    ''') % REPO
    with tempfile.TemporaryDirectory() as tmp:
        res = subprocess.run([sys.executable, '-c', script], cwd=tmp,
                             capture_output=True, text=True, timeout=300)
        out = res.stdout
        assert 'TRAIN_DONE' in out, (out[-3000:], res.stderr[-3000:])
        assert 'started gpu actor pool' in out
        assert 'updated model(' in out
        assert os.path.exists(os.path.join(tmp, 'models', '1.pth'))


def test_train_with_device_replay_solo():
    """--train with replay: 'device' (HungryGeese solo): the HBM-ring
    replay replaces the episode buffer + batchers; eager gather-train on
    CPU here (captured on GPU)."""
    script = textwrap.dedent('''
        import sys
        sys.path.insert(0, %r)
        from handyrl_amd.train import train_main
        args = {
            'env_args': {'env': 'HungryGeese'},
            'train_args': {
                'turn_based_training': False, 'observation': False,
                'gamma': 0.8, 'forward_steps': 8, 'burn_in_steps': 0,
                'compress_steps': 4, 'entropy_regularization': 0.1,
                'entropy_regularization_decay': 0.1, 'update_episodes': 30,
                'batch_size': 4, 'minimum_episodes': 10, 'maximum_episodes': 300,
                'epochs': 1, 'num_batchers': 1, 'eval_rate': 0.1,
                'worker': {'type': 'gpu', 'num_parallel': 1, 'num_envs': 8},
                'lambda': 0.7, 'policy_target': 'VTRACE', 'value_target': 'VTRACE',
                'eval': {'opponent': ['random']}, 'seed': 0, 'restart_epoch': 0,
                'bf16': False, 'replay': 'device',
            },
        }
        train_main(args)
        print('TRAIN_DONE')
    ''') % REPO
    with tempfile.TemporaryDirectory() as tmp:
        res = subprocess.run([sys.executable, '-c', script], cwd=tmp,
                             capture_output=True, text=True, timeout=300)
        out = res.stdout
        assert 'TRAIN_DONE' in out, (out[-3000:], res.stderr[-3000:])
        assert 'updated model(' in out
        assert os.path.exists(os.path.join(tmp, 'models', '1.pth'))


def test_train_with_device_replay_turn_based():
    """--train with replay: 'device' on the recurrent Geister config,
    including a burn-in prefix (device-gather lead pads)."""
    script = textwrap.dedent('''
        import sys
        sys.path.insert(0, %r)
        from handyrl_amd.train import train_main
        args = {
            'env_args': {'env': 'Geister'},
            'train_args': {
                'turn_based_training': True, 'observation': False,
                'gamma': 0.8, 'forward_steps': 8, 'burn_in_steps': 2,
                'compress_steps': 4, 'entropy_regularization': 0.1,
                'entropy_regularization_decay': 0.1, 'update_episodes': 20,
                'batch_size': 2, 'minimum_episodes': 4, 'maximum_episodes': 200,
                'epochs': 1, 'num_batchers': 1, 'eval_rate': 0.1,
                'worker': {'type': 'gpu', 'num_parallel': 1, 'num_envs': 8},
                'lambda': 0.7, 'policy_target': 'UPGO', 'value_target': 'TD',
                'eval': {'opponent': ['random']}, 'seed': 0, 'restart_epoch': 0,
                'bf16': False, 'replay': 'device',
            },
        }
        train_main(args)
        print('TRAIN_DONE')
    ''') % REPO
    with tempfile.TemporaryDirectory() as tmp:
        res = subprocess.run([sys.executable, '-c', script], cwd=tmp,
                             capture_output=True, text=True, timeout=300)
        out = res.stdout
        assert 'TRAIN_DONE' in out, (out[-3000:], res.stderr[-3000:])
        assert 'updated model(' in out
        assert os.path.exists(os.path.join(tmp, 'models', '1.pth'))


def test_device_replay_config_validation():
    """replay: 'device' rejects unsupported configs with clear errors."""
    import pytest as _pytest
    import torch
    from handyrl_amd.train import Trainer
    from handyrl_amd.envs.tictactoe import SimpleConv2dModel
    base = {
        'turn_based_training': True, 'observation': True, 'gamma': 0.8,
        'forward_steps': 4, 'burn_in_steps': 0, 'compress_steps': 4,
        'entropy_regularization': 0.1, 'entropy_regularization_decay': 0.1,
        'batch_size': 2, 'minimum_episodes': 1, 'maximum_episodes': 10,
        'num_batchers': 1, 'lambda': 0.7, 'policy_target': 'TD',
        'value_target': 'TD', 'seed': 0, 'bf16': False,
        'replay': 'device',
    }
    with _pytest.raises(ValueError, match='observation'):
        Trainer(dict(base), SimpleConv2dModel(), device=torch.device('cpu'))
    bad = dict(base)
    bad['turn_based_training'] = False
    bad['observation'] = False
    bad['burn_in_steps'] = 2
    with _pytest.raises(ValueError, match='burn_in'):
        Trainer(bad, SimpleConv2dModel(), device=torch.device('cpu'))


def test_gpu_actor_mode_respects_epoch_limit():
    """GPU actors buffer many episodes; the backlog must not re-fire
    epoch rollovers past the configured count during the shutdown drain
    (regression: epochs=1 once rolled to 7 on hardware)."""
    script = textwrap.dedent('''
        import sys
        sys.path.insert(0, %r)
        from handyrl_amd.train import train_main
        args = {
            'env_args': {'env': 'HungryGeese'},
            'train_args': {
                'turn_based_training': False, 'observation': False,
                'gamma': 0.8, 'forward_steps': 8, 'burn_in_steps': 0,
                'compress_steps': 4, 'entropy_regularization': 0.1,
                'entropy_regularization_decay': 0.1, 'update_episodes': 10,
                'batch_size': 4, 'minimum_episodes': 10,
                'maximum_episodes': 300,
                'epochs': 1, 'num_batchers': 1, 'eval_rate': 0.1,
                'worker': {'type': 'gpu', 'num_parallel': 0, 'num_envs': 64},
                'lambda': 0.7, 'policy_target': 'VTRACE',
                'value_target': 'VTRACE',
                'eval': {'opponent': ['random']}, 'seed': 0,
                'restart_epoch': 0, 'bf16': False,
            },
        }
        train_main(args)
        print('TRAIN_DONE')
    ''') % REPO
    with tempfile.TemporaryDirectory() as tmp:
        res = subprocess.run([sys.executable, '-c', script], cwd=tmp,
                             capture_output=True, text=True, timeout=300)
        out = res.stdout
        assert 'TRAIN_DONE' in out, (out[-3000:], res.stderr[-3000:])
        # 64 envs with update_episodes=10 buffer a large backlog; exactly
        # ONE epoch must be written
        assert os.path.exists(os.path.join(tmp, 'models', '1.pth'))
        assert not os.path.exists(os.path.join(tmp, 'models', '2.pth')), out
