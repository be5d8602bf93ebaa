"""Distributed-learner tests over gloo (world_size 2, CPU) — the DP path
that runs over RCCL/xGMI on an MI355X node, exercised here with the same
code and a CPU backend."""

import os
import subprocess
import sys
import textwrap

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

WORKER = textwrap.dedent('''
    import os, sys
    sys.path.insert(0, %r)
    import torch
    import torch.distributed as dist
    from handyrl_amd import dist as hdist
    from handyrl_amd.dist import GradReducer, allreduce_scalar, broadcast_params

    rank = int(os.environ['RANK'])
    dist.init_process_group('gloo')

    torch.manual_seed(100 + rank)                 # deliberately different
    model = torch.nn.Sequential(torch.nn.Linear(4, 8), torch.nn.Linear(8, 2))

    # 1) parameter broadcast aligns replicas
    broadcast_params(model, src=0)
    flat = torch.cat([p.flatten() for p in model.parameters()])
    gathered = [torch.empty_like(flat) for _ in range(2)]
    dist.all_gather(gathered, flat)
    assert torch.equal(gathered[0], gathered[1]), 'broadcast failed'

    # 2) gradient all-reduce sums across ranks
    x = torch.full((3, 4), float(rank + 1))
    loss = model(x).sum()
    loss.backward()
    local_grads = [p.grad.clone() for p in model.parameters()]
    summed = [g.clone() for g in local_grads]
    for g in summed:
        dist.all_reduce(g)                         # oracle
    for p, lg in zip(model.parameters(), local_grads):
        p.grad = lg.clone()
    reducer = GradReducer(model.parameters())
    reducer.allreduce_()
    for p, expect in zip(model.parameters(), summed):
        assert torch.allclose(p.grad, expect), 'grad allreduce mismatch'

    # 3) scalar data-count reduce
    total = allreduce_scalar(10.0 * (rank + 1))
    assert abs(total - 30.0) < 1e-6

    print('DIST_OK rank', rank)
''') % REPO


def test_gloo_dp_primitives():
    env = dict(os.environ)
    env.update({'MASTER_ADDR': '127.0.0.1', 'MASTER_PORT': '29541',
                'WORLD_SIZE': '2', 'GLOO_SOCKET_IFNAME': 'lo'})
    procs = []
    for rank in range(2):
        e = dict(env, RANK=str(rank))
        procs.append(subprocess.Popen([sys.executable, '-c', WORKER], env=e,
                                      stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
                                      text=True))
    outs = []
    for p in procs:
        out, _ = p.communicate(timeout=120)
        outs.append(out)
    for rank, (p, out) in enumerate(zip(procs, outs)):
        assert p.returncode == 0, (rank, out[-3000:])
        assert 'DIST_OK' in out


TRAIN_WORKER = textwrap.dedent('''
    import os, random, sys
    sys.path.insert(0, %r)
    import torch
    import torch.distributed as dist
    from handyrl_amd.batch import make_batch, EpisodeBuffer
    from handyrl_amd.generation import Generator
    from handyrl_amd.model import ModelWrapper
    from handyrl_amd.train import Trainer
    from handyrl_amd.dist import broadcast_params
    from handyrl_amd.envs import tictactoe

    rank = int(os.environ['RANK'])
    dist.init_process_group('gloo')

    args = {
        'turn_based_training': True, 'observation': False, 'gamma': 0.8,
        'forward_steps': 4, 'burn_in_steps': 0, 'compress_steps': 4,
        'entropy_regularization': 0.1, 'entropy_regularization_decay': 0.1,
        'batch_size': 4, 'minimum_episodes': 2, 'maximum_episodes': 100,
        'num_batchers': 1, 'lambda': 0.7, 'policy_target': 'TD',
        'value_target': 'TD', 'seed': 0, 'bf16': False,
    }
    env = tictactoe.Environment()
    net = env.net()
    trainer = Trainer(args, net, device=torch.device('cpu'))
    broadcast_params(trainer.model, src=0)

    # per-rank distinct self-play data (the DP weak-scaling shape)
    random.seed(1000 + rank)
    gen = Generator(tictactoe.Environment(), args)
    models = {p: ModelWrapper(env.net()) for p in env.players()}
    job = {'player': env.players(), 'model_id': {p: 1 for p in env.players()}}
    buf = EpisodeBuffer(args)
    buf.extend([gen.generate(models, job) for _ in range(6)])

    for step in range(3):
        batch = make_batch([buf.select_episode() for _ in range(args['batch_size'])], args)
        losses, dcnt = trainer.train_step(batch)
        assert torch.isfinite(losses['total'])

    # replicas must stay bit-identical after synced steps
    flat = torch.cat([p.detach().flatten() for p in trainer.params])
    gathered = [torch.empty_like(flat) for _ in range(2)]
    dist.all_gather(gathered, flat)
    assert torch.equal(gathered[0], gathered[1]), 'replicas diverged'
    print('DP_TRAIN_OK rank', rank)
''') % REPO


def test_gloo_dp_training_replicas_stay_synced():
    env = dict(os.environ)
    env.update({'MASTER_ADDR': '127.0.0.1', 'MASTER_PORT': '29542',
                'WORLD_SIZE': '2', 'GLOO_SOCKET_IFNAME': 'lo'})
    procs = []
    for rank in range(2):
        e = dict(env, RANK=str(rank))
        procs.append(subprocess.Popen([sys.executable, '-c', TRAIN_WORKER], env=e,
                                      stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
                                      text=True))
    for rank, p in enumerate(procs):
        out, _ = p.communicate(timeout=240)
        assert p.returncode == 0, (rank, out[-3000:])
        assert 'DP_TRAIN_OK' in out


DP_EQUIV = textwrap.dedent('''
    import os, pickle, sys
    sys.path.insert(0, %r)
    import random
    import torch
    import torch.distributed as dist
    from handyrl_amd.batch import make_batch, EpisodeBuffer
    from handyrl_amd.generation import Generator
    from handyrl_amd.model import ModelWrapper
    from handyrl_amd.train import Trainer
    from handyrl_amd.envs import tictactoe

    rank = int(os.environ['RANK'])
    dist.init_process_group('gloo')
    args = {
        'turn_based_training': True, 'observation': False, 'gamma': 0.8,
        'forward_steps': 8, 'burn_in_steps': 0, 'compress_steps': 4,
        'entropy_regularization': 0.1, 'entropy_regularization_decay': 0.1,
        'batch_size': 4, 'minimum_episodes': 2, 'maximum_episodes': 100,
        'num_batchers': 1, 'lambda': 0.7, 'policy_target': 'TD',
        'value_target': 'TD', 'seed': 0, 'bf16': False,
        'compress_episodes': False,
    }
    env = tictactoe.Environment()
    torch.manual_seed(0)
    net = env.net()

    # identical episode pool on both ranks
    gen = Generator(env, args)
    models = {p: ModelWrapper(net) for p in env.players()}
    job = {'player': env.players(), 'model_id': {p: 1 for p in env.players()}}
    random.seed(7)
    eps = [gen.generate(models, job) for _ in range(6)]
    buf = EpisodeBuffer(args)
    buf.extend(eps)
    random.seed(11)
    sels = [buf.select_episode() for _ in range(8)]

    # single-process oracle on the FULL batch of 8 windows (rank 0 only)
    import copy
    from handyrl_amd.train import compute_loss
    if rank == 0:
        net_full = copy.deepcopy(net)
        full = make_batch(sels, args)
        losses, dcnt = compute_loss(full, ModelWrapper(net_full), None, args)
        losses['total'].backward()
        oracle = [p.grad.clone() for p in net_full.parameters()]

    # DP: each rank computes loss/backward on HALF the windows, then the
    # fused all-reduce sums gradients (pre-clip comparison point)
    mine = sels[rank * 4:(rank + 1) * 4]
    batch = make_batch(mine, args)
    trainer2 = Trainer(args, copy.deepcopy(net), device=torch.device('cpu'),
                       batcher=False)
    l2, _ = compute_loss(batch, trainer2.wrapped_model, None, args)
    trainer2.optimizer.zero_grad(set_to_none=False)
    l2['total'].backward()
    trainer2.reducer.allreduce_()
    if rank == 0:
        for p, og in zip(trainer2.model.parameters(), oracle):
            assert torch.allclose(p.grad, og, rtol=1e-5, atol=1e-6), \
                'DP summed grads != single-process full-batch grads'
        print('DP_EQUIV_OK')
    dist.barrier()
''') % REPO


def test_dp_grads_equal_single_process():
    """Two DP ranks on half-batches produce EXACTLY the single-process
    full-batch gradients (the reference's summed-loss DataParallel
    semantics, reference train.py:339-371)."""
    env = dict(os.environ)
    env.update({'MASTER_ADDR': '127.0.0.1', 'MASTER_PORT': '29557',
                'WORLD_SIZE': '2'})
    procs = []
    for r in range(2):
        e = dict(env)
        e['RANK'] = str(r)
        procs.append(subprocess.Popen(
            [sys.executable, '-c', DP_EQUIV], env=e,
            stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True))
    outs = [p.communicate(timeout=240)[0] for p in procs]
    assert all(p.returncode == 0 for p in procs), outs
    assert 'DP_EQUIV_OK' in outs[0], outs[0][-2000:]
