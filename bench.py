"""Flagship benchmark: Hungry Geese self-play training throughput.

Measures the BASELINE.json headline metric (learner steps/sec + env
frames/sec, Hungry Geese self-play) on N GPUs of one node: each rank runs
the full pipeline — vectorized self-play with batched bf16 GPU inference
(handyrl_amd/actor.py), parallel batch builders, and V-Trace learner steps
with fused HIP target scans — with gradients all-reduced over RCCL/xGMI
(one process per GPU; weak scaling: per-GPU work is fixed).

Each timed step is a fixed work quantum:
  ACTOR_VEC_STEPS vectorized env transitions over N_ENVS games (self-play
  generation feeding the replay buffer) + exactly one optimizer step on a
  (batch_size x forward_steps) V-Trace batch.

Launch (driver contract):
  python bench.py --gpus 1 --steps K --warmup W
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W
"""

import argparse
import json
import os
import time

import torch

from handyrl_amd import dist as hdist
from handyrl_amd.actor import GeeseActorPool, PipelinedGeesePool
from handyrl_amd.batch import Batcher
from handyrl_amd.models.geese_net import GeeseNet
from handyrl_amd.train import Trainer

N_ENVS = 4096           # self-play games per GPU (12 env-worker processes)
ACTOR_VEC_STEPS = 16    # env transitions (per game) per learner step


def bench_args(batch_size=128, forward_steps=16):
    return {
        'turn_based_training': False,     # 4-player simultaneous: solo seats
        'observation': False,
        'gamma': 0.8,
        'forward_steps': forward_steps,
        'burn_in_steps': 0,
        'compress_steps': 4,
        'entropy_regularization': 0.1,
        'entropy_regularization_decay': 0.1,
        'batch_size': batch_size,
        'minimum_episodes': 160,
        'maximum_episodes': 4000,
        'num_batchers': 3,
        'lambda': 0.7,
        'policy_target': 'VTRACE',
        'value_target': 'VTRACE',
        'seed': 0,
        'bf16': True,
        'compress_episodes': False,       # local buffer: skip bz2
    }


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument('--gpus', type=int, default=1)
    parser.add_argument('--steps', type=int, default=30)
    # self-play throughput is policy-dependent: random-init episodes are
    # short (heavy reset/package load) and lengthen as the policy learns
    # during the run.  A longer default warmup lets the measured window
    # start in the learned (steady-state) regime.
    parser.add_argument('--warmup', type=int, default=30)
    parser.add_argument('--envs', type=int, default=N_ENVS)
    parser.add_argument('--batch-size', type=int, default=128)
    parser.add_argument('--forward-steps', type=int, default=16)
    cli = parser.parse_args()

    # single-threaded CPU torch: the GPU does the math, and forked batch
    # builders deadlock if the parent has a live OpenMP pool at fork time
    torch.set_num_threads(1)
    rank = hdist.env_rank()
    local_rank = hdist.env_local_rank()
    world = hdist.env_world_size()

    args = bench_args(cli.batch_size, cli.forward_steps)

    # fork the batch-builder and actor-env processes FIRST, before any HIP
    # context or torch thread-pool exists in this process
    from handyrl_amd.batch import EpisodeBuffer
    buffer = EpisodeBuffer(args)
    # with the device-resident replay the batch is gathered ON DEVICE inside
    # the training graph: no batch-builder processes at all
    device_replay = torch.cuda.is_available() and \
        os.environ.get('HANDYRL_DEVICE_REPLAY', '1') == '1'
    batcher = False if device_replay else Batcher(args, buffer)

    # measured round-2 defaults (interleaved sweeps, profiles/b9_*):
    # envs=4096 + 12 workers gives 752-812k frames/s at ~12 learner
    # steps/s on one MI355X; multi-rank runs keep 8 workers per GPU so an
    # 8-GPU node stays within its core budget
    actor_procs = int(os.environ.get(
        'HANDYRL_ACTOR_PROCS',
        '12' if hdist.env_world_size() == 1 else '8'))
    # single slot per worker: 2-slot double-buffering measured SLOWER
    # end-to-end (it steers self-play into the short-episode regime;
    # BASELINE.md post-fix slots comparison)
    os.environ.setdefault('HANDYRL_ACTOR_SLOTS', '1')
    # device-side trajectory recording (handyrl_amd/traj): the actor graph
    # scatters obs/alive/(action,prob,value) into HBM rings; finished
    # episodes commit D2D into the replay ring — env workers ship only
    # metadata, never observation arrays.  HANDYRL_TRAJ=0 restores the
    # host-recorded episode path.
    traj_mode = torch.cuda.is_available() and device_replay and \
        os.environ.get('HANDYRL_TRAJ', '1') == '1'
    mpool = None
    if actor_procs > 0:
        from handyrl_amd.actor import MultiProcGeesePool
        mpool = MultiProcGeesePool(args, n_games=cli.envs,
                                   seed=1000 + hdist.env_rank() * 31,
                                   workers=actor_procs,
                                   traj_mode=traj_mode,
                                   make_stubs=not traj_mode)

    use_cuda = torch.cuda.is_available()
    device = torch.device('cuda', local_rank) if use_cuda else torch.device('cpu')
    if use_cuda:
        torch.cuda.set_device(device)
    hdist.init_from_env(device=local_rank if use_cuda else None)

    torch.manual_seed(1234)                      # identical init on all ranks
    model = GeeseNet()

    trainer = Trainer(args, model, device=device, episodes=buffer, batcher=batcher)
    if world > 1:
        hdist.broadcast_params(trainer.model)

    replay = None
    if device_replay:
        from handyrl_amd.replay import DeviceReplay
        replay = DeviceReplay(args, device, bytes_budget=int(
            os.environ.get('HANDYRL_REPLAY_BYTES', str(4 << 30))),
            ingest_thread=True)

    if mpool is not None:
        # env work in child processes; parent runs the inference engine
        mpool.attach(trainer.model, device, replay=replay)
        pool = mpool
    else:
        # in-process fallback: pipelined two-shard pool on GPU, plain on CPU
        pool_cls = PipelinedGeesePool if use_cuda else GeeseActorPool
        pool = pool_cls(trainer.model, args, n_games=cli.envs,
                        device=device, seed=1000 + rank)
    actor_calls = getattr(pool, 'calls_per_vec_step', 1)

    def pump_actor(n_vec_steps):
        frames = 0
        was_training = trainer.model.training
        trainer.model.eval()
        for _ in range(n_vec_steps * actor_calls):
            frames += pool.step_once()
        if was_training:
            trainer.model.train()
        eps = pool.harvest()
        if eps:
            (replay if replay is not None else trainer.episodes).extend(eps)
        return frames

    # ---- prefill: generate the minimum episode set (untimed) ----
    t0 = time.time()
    while pool.episodes_done < args['minimum_episodes']:
        pump_actor(8)
    trainer.episodes.trim(args['maximum_episodes'])
    if batcher:
        batcher.run()
    if rank == 0:
        import sys
        print('# prefill: %d episodes in %.1fs' %
              (len(trainer.episodes), time.time() - t0), file=sys.stderr, flush=True)

    phase_t = {'actor': 0.0, 'batch_wait': 0.0, 'train': 0.0, 'n': 0}

    replay_step = [None]

    def one_step():
        t0 = time.time()
        frames = pump_actor(ACTOR_VEC_STEPS)
        t1 = time.time()
        if replay_step[0] is not None:
            t2 = time.time()
            # sample-gather + train, one graph replay, issued async
            losses, _ = replay_step[0].step()
        else:
            batch = batcher.batch()
            t2 = time.time()
            graphed = getattr(trainer, 'graphed_step', None)
            if graphed is not None:
                # async issue: the GPU time hides under the next actor phase
                losses, _ = graphed.step(batch)
            else:
                losses, dcnt = trainer.train_step(batch)
        pool.refresh_weights()          # re-fold BN into the MFMA actor path
        t3 = time.time()
        phase_t['actor'] += t1 - t0
        phase_t['batch_wait'] += t2 - t1
        phase_t['train'] += t3 - t2
        phase_t['n'] += 1
        return frames, losses

    # ---- capture the train step as a hipGraph (fixed shapes) ----
    if replay is not None:
        from handyrl_amd.hipgraph import GraphedReplayTrainStep
        replay.flush()                 # ensure the prefill block is published
        replay_step[0] = GraphedReplayTrainStep(trainer, replay, cli.batch_size)
        if rank == 0:
            import sys
            print('# replay train-step hipGraph: %s' %
                  ('captured' if replay_step[0].graph is not None else 'EAGER'),
                  file=sys.stderr, flush=True)
    elif use_cuda and os.environ.get('HANDYRL_NO_GRAPHS') != '1':
        example = batcher.batch()
        ok = trainer.enable_cuda_graph(example)
        if rank == 0:
            import sys
            print('# train-step hipGraph: %s' % ('captured' if ok else 'EAGER'),
                  file=sys.stderr, flush=True)

    # ---- warmup (untimed) ----
    for _ in range(cli.warmup):
        one_step()

    # ---- timed region ----
    hdist.barrier()
    if use_cuda:
        torch.cuda.synchronize()
    start = time.time()
    total_frames = 0
    for _ in range(cli.steps):
        frames, _ = one_step()
        total_frames += frames
    if use_cuda:
        torch.cuda.synchronize()
    hdist.barrier()
    elapsed = time.time() - start

    # max over ranks (slowest rank defines job time)
    if world > 1:
        import torch.distributed as tdist
        t = torch.tensor([elapsed], device=device if use_cuda else 'cpu')
        tdist.all_reduce(t, op=tdist.ReduceOp.MAX)
        elapsed = t.item()
        f = torch.tensor([float(total_frames)], device=device if use_cuda else 'cpu')
        tdist.all_reduce(f, op=tdist.ReduceOp.SUM)
        total_frames = int(f.item())

    n_gpus = world if world > 1 else cli.gpus
    steps_per_sec = cli.steps / elapsed
    frames_per_sec = total_frames / elapsed
    samples_per_sec = steps_per_sec * cli.batch_size * cli.forward_steps * n_gpus

    if rank == 0:
        import sys
        n = max(1, phase_t['n'])
        print('# phase ms/step: actor=%.1f batch_wait=%.1f train=%.1f' %
              (1000 * phase_t['actor'] / n, 1000 * phase_t['batch_wait'] / n,
               1000 * phase_t['train'] / n), file=sys.stderr, flush=True)
        tm = pool.timing
        an = max(1, tm['n'])
        print('# actor ms/vecstep: ' + ' '.join(
            '%s=%.2f' % (k, 1000 * tm[k] / an) for k in
            ('obs', 'fwd', 'sample', 'record', 'env', 'package')),
            file=sys.stderr, flush=True)
        eps = max(1, getattr(pool, 'episodes_done', 0))
        print('# episodes=%d mean_len=%.1f (regime check: random-init ~8, '
              'learned 50-80)' % (eps, getattr(pool, 'frames', 0) / eps),
              file=sys.stderr, flush=True)
        result = {
            'metric': 'hungry_geese_selfplay_env_frames_per_sec',
            'value': round(frames_per_sec, 1),
            'unit': 'frames/s',
            'n_gpus': n_gpus,
            'steps': cli.steps,
            'warmup': cli.warmup,
            'ms_per_step': round(1000.0 * elapsed / cli.steps, 2),
            'higher_is_better': True,
            'scaling': 'weak',
            'vs_baseline': None,
            'dtype': 'bf16' if (args['bf16'] and use_cuda) else 'fp32',
            'data': 'synthetic',
            'config': {
                'model': 'GeeseNet(12x32 torus-conv residual)',
                'global_batch': cli.batch_size * n_gpus,
                'seq_len': cli.forward_steps,
                'parallelism': 'dp%d' % n_gpus,
                'envs_per_gpu': cli.envs,
                'actor_vec_steps_per_learner_step': ACTOR_VEC_STEPS,
                'learner_steps_per_sec': round(steps_per_sec, 2),
                'learner_samples_per_sec': round(samples_per_sec, 1),
                'loss': 'VTRACE (policy+value), entropy reg',
                'note': 'self-play on random-init weights; no external data',
            },
        }
        print(json.dumps(result), flush=True)

    if hasattr(pool, 'shutdown'):
        pool.shutdown()


if __name__ == '__main__':
    main()
