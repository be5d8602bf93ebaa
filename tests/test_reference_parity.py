"""Numerical parity against the ACTUAL reference implementation: the same
batch through our compute_loss and DeNA/HandyRL's compute_loss (imported
read-only from /root/reference) must produce identical losses.

Skipped automatically when the reference tree is not mounted.
"""

import copy
import os
import random
import sys

import pytest
import torch

REFERENCE = '/root/reference'
pytestmark = pytest.mark.skipif(
    not os.path.isdir(os.path.join(REFERENCE, 'handyrl')),
    reason='reference tree not available')


def _args(**over):
    args = {
        'turn_based_training': True, 'observation': False, 'gamma': 0.8,
        'forward_steps': 8, 'burn_in_steps': 0, 'compress_steps': 4,
        'entropy_regularization': 0.1, 'entropy_regularization_decay': 0.1,
        'batch_size': 6, 'minimum_episodes': 2, 'maximum_episodes': 100,
        'num_batchers': 1, 'lambda': 0.7, 'policy_target': 'VTRACE',
        'value_target': 'TD', 'seed': 0, 'bf16': False,
        'compress_episodes': True,
    }
    args.update(over)
    return args


@pytest.mark.parametrize('policy_target,value_target', [
    ('TD', 'TD'), ('VTRACE', 'VTRACE'), ('UPGO', 'TD'), ('MC', 'MC'),
    ('VTRACE', 'UPGO'),
])
def test_compute_loss_matches_reference(policy_target, value_target):
    sys.path.insert(0, REFERENCE)
    try:
        from handyrl.train import compute_loss as ref_compute_loss
        from handyrl.model import ModelWrapper as RefWrapper
    finally:
        sys.path.remove(REFERENCE)

    from handyrl_amd.batch import make_batch, EpisodeBuffer
    from handyrl_amd.generation import Generator
    from handyrl_amd.model import ModelWrapper
    from handyrl_amd.train import compute_loss
    from handyrl_amd.envs import tictactoe

    args = _args(policy_target=policy_target, value_target=value_target)
    env = tictactoe.Environment()
    torch.manual_seed(0)
    net = env.net()
    gen = Generator(env, args)
    models = {p: ModelWrapper(copy.deepcopy(net)) for p in env.players()}
    job = {'player': env.players(), 'model_id': {p: 1 for p in env.players()}}
    buf = EpisodeBuffer(args)
    random.seed(3)
    buf.extend([gen.generate(models, job) for _ in range(5)])
    batch = make_batch([buf.select_episode() for _ in range(args['batch_size'])],
                       args)

    net_ours = copy.deepcopy(net)
    net_ref = copy.deepcopy(net)
    torch.manual_seed(1)
    losses_ours, dcnt_ours = compute_loss(batch, ModelWrapper(net_ours), None, args)
    torch.manual_seed(1)
    losses_ref, dcnt_ref = ref_compute_loss(
        {k: v.clone() for k, v in batch.items()}, RefWrapper(net_ref), None, args)

    assert float(dcnt_ours) == pytest.approx(float(dcnt_ref))
    for key in losses_ref:
        assert key in losses_ours, key
        torch.testing.assert_close(
            losses_ours[key].double(), losses_ref[key].double(),
            rtol=1e-4, atol=1e-5,
            msg=lambda m, key=key: '%s: %s' % (key, m))

    # gradients through the whole graph must match too
    losses_ours['total'].backward()
    losses_ref['total'].backward()
    for (n, p_o), (_, p_r) in zip(net_ours.named_parameters(),
                                  net_ref.named_parameters()):
        torch.testing.assert_close(p_o.grad, p_r.grad, rtol=1e-4, atol=1e-6,
                                   msg=lambda m, n=n: '%s: %s' % (n, m))


def test_make_batch_matches_reference():
    """Identical episode windows through our make_batch and the
    reference's must produce identical tensors (turn-based path)."""
    sys.path.insert(0, REFERENCE)
    try:
        from handyrl.train import make_batch as ref_make_batch
    finally:
        sys.path.remove(REFERENCE)

    from handyrl_amd.batch import make_batch, EpisodeBuffer
    from handyrl_amd.generation import Generator
    from handyrl_amd.model import ModelWrapper
    from handyrl_amd.envs import tictactoe

    args = _args(forward_steps=6, burn_in_steps=0)
    env = tictactoe.Environment()
    torch.manual_seed(0)
    gen = Generator(env, args)
    models = {p: ModelWrapper(env.net()) for p in env.players()}
    job = {'player': env.players(), 'model_id': {p: 1 for p in env.players()}}
    buf = EpisodeBuffer(args)
    random.seed(9)
    buf.extend([gen.generate(models, job) for _ in range(4)])
    sels = [buf.select_episode() for _ in range(5)]

    ours = make_batch([dict(s) for s in sels], args)
    theirs = ref_make_batch([dict(s) for s in sels], args)
    assert set(ours.keys()) == set(theirs.keys())
    for key in theirs:
        torch.testing.assert_close(ours[key].double(), theirs[key].double(),
                                   rtol=0, atol=0,
                                   msg=lambda m, key=key: '%s: %s' % (key, m))


def test_recurrent_compute_loss_matches_reference():
    """RNN path parity: Geister episodes (DRC hidden state, turn-based,
    partial observation) through our compute_loss/forward_prediction and
    the reference's own — identical losses and gradients."""
    sys.path.insert(0, REFERENCE)
    try:
        from handyrl.train import compute_loss as ref_compute_loss
        from handyrl.model import ModelWrapper as RefWrapper
    finally:
        sys.path.remove(REFERENCE)

    from handyrl_amd.batch import make_batch, EpisodeBuffer
    from handyrl_amd.generation import Generator
    from handyrl_amd.model import ModelWrapper
    from handyrl_amd.train import compute_loss
    from handyrl_amd.envs import geister

    args = _args(policy_target='UPGO', value_target='TD',
                 forward_steps=6, burn_in_steps=2)
    env = geister.Environment()
    torch.manual_seed(4)
    net = env.net()
    gen = Generator(env, args)
    models = {p: ModelWrapper(copy.deepcopy(net)) for p in env.players()}
    job = {'player': env.players(), 'model_id': {p: 1 for p in env.players()}}
    buf = EpisodeBuffer(args)
    random.seed(9)
    buf.extend([gen.generate(models, job) for _ in range(4)])
    B = 4
    batch = make_batch([buf.select_episode() for _ in range(B)], args)

    net_ours = copy.deepcopy(net)
    net_ref = copy.deepcopy(net)
    wrap_ours = ModelWrapper(net_ours)
    wrap_ref = RefWrapper(net_ref)
    P = batch['value'].size(2)
    torch.manual_seed(2)
    losses_ours, dcnt_ours = compute_loss(
        batch, wrap_ours, wrap_ours.init_hidden([B, P]), args)
    torch.manual_seed(2)
    losses_ref, dcnt_ref = ref_compute_loss(
        {k: (v.clone() if torch.is_tensor(v) else
             {kk: vv.clone() for kk, vv in v.items()})
         for k, v in batch.items()},
        wrap_ref, wrap_ref.init_hidden([B, P]), args)

    assert float(dcnt_ours) == pytest.approx(float(dcnt_ref))
    for key in losses_ref:
        torch.testing.assert_close(
            losses_ours[key].double(), losses_ref[key].double(),
            rtol=1e-4, atol=1e-5,
            msg=lambda m, key=key: '%s: %s' % (key, m))
    losses_ours['total'].backward()
    losses_ref['total'].backward()
    for (n, p_o), (_, p_r) in zip(net_ours.named_parameters(),
                                  net_ref.named_parameters()):
        if p_o.grad is None and p_r.grad is None:
            continue
        torch.testing.assert_close(p_o.grad, p_r.grad, rtol=1e-4, atol=1e-6,
                                   msg=lambda m, n=n: '%s: %s' % (n, m))


@pytest.mark.parametrize('env_name,over', [
    ('parallel_tictactoe', {'turn_based_training': False}),   # simultaneous
    ('tictactoe', {'observation': True}),                     # observer rows
])
def test_other_training_families_match_reference(env_name, over):
    """Simultaneous-move (ParallelTicTacToe) and observation=True
    (observer-trained) configs produce reference-identical losses."""
    sys.path.insert(0, REFERENCE)
    try:
        from handyrl.train import compute_loss as ref_compute_loss
        from handyrl.model import ModelWrapper as RefWrapper
    finally:
        sys.path.remove(REFERENCE)

    from handyrl_amd.batch import make_batch, EpisodeBuffer
    from handyrl_amd.generation import Generator
    from handyrl_amd.model import ModelWrapper
    from handyrl_amd.train import compute_loss
    import importlib
    env_mod = importlib.import_module('handyrl_amd.envs.%s' % env_name)

    args = _args(policy_target='TD', value_target='TD', **over)
    env = env_mod.Environment()
    torch.manual_seed(7)
    net = env.net()
    gen = Generator(env, args)
    models = {p: ModelWrapper(copy.deepcopy(net)) for p in env.players()}
    job = {'player': env.players(), 'model_id': {p: 1 for p in env.players()}}
    buf = EpisodeBuffer(args)
    random.seed(17)
    buf.extend([gen.generate(models, job) for _ in range(5)])
    B = 4
    batch = make_batch([buf.select_episode() for _ in range(B)], args)

    net_ours = copy.deepcopy(net)
    net_ref = copy.deepcopy(net)
    torch.manual_seed(3)
    losses_ours, dcnt_ours = compute_loss(batch, ModelWrapper(net_ours),
                                          None, args)
    torch.manual_seed(3)
    losses_ref, dcnt_ref = ref_compute_loss(
        {k: v.clone() for k, v in batch.items()}, RefWrapper(net_ref),
        None, args)
    assert float(dcnt_ours) == pytest.approx(float(dcnt_ref))
    for key in losses_ref:
        torch.testing.assert_close(
            losses_ours[key].double(), losses_ref[key].double(),
            rtol=1e-4, atol=1e-5,
            msg=lambda m, key=key: '%s: %s' % (key, m))
