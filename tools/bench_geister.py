# Geister self-play throughput (BASELINE.json config #5 shape): N actors
# in env-worker processes + batched recurrent (DRC) GPU inference, with a
# concurrent RNN learner (UPGO+TD, bf16).  Prints one JSON line.
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    import argparse
    ap = argparse.ArgumentParser()
    ap.add_argument('--actors', type=int, default=256)
    ap.add_argument('--workers', type=int, default=8)
    ap.add_argument('--steps', type=int, default=30)
    ap.add_argument('--warmup', type=int, default=5)
    ap.add_argument('--batch-size', type=int, default=32)
    ap.add_argument('--device-replay', dest='device_replay',
                    action='store_true', default=True,
                    help='HBM-resident TurnDeviceReplay + captured '
                         'recurrent train step (DEFAULT: measured 1.75x '
                         'the host batcher path, BASELINE.md round 2)')
    ap.add_argument('--host-replay', dest='device_replay',
                    action='store_false',
                    help='host EpisodeBuffer + multiprocess Batcher path')
    cli = ap.parse_args()

    torch.set_num_threads(1)
    args = {
        'turn_based_training': True, 'observation': False, 'gamma': 0.8,
        'forward_steps': 16, 'burn_in_steps': 0, 'compress_steps': 4,
        'entropy_regularization': 0.1, 'entropy_regularization_decay': 0.1,
        'batch_size': cli.batch_size, 'minimum_episodes': 80,
        'maximum_episodes': 2000, 'num_batchers': 3, 'lambda': 0.7,
        'policy_target': 'UPGO', 'value_target': 'TD', 'seed': 0,
        'bf16': False,                    # RNN path trains fp32 (small net)
        'compress_episodes': False,
    }

    from handyrl_amd.batch import EpisodeBuffer, Batcher
    buffer = EpisodeBuffer(args)
    batcher = None if cli.device_replay else Batcher(args, buffer)
    import torch as _torch
    traj_mode = cli.device_replay and _torch.cuda.is_available() and \
        os.environ.get('HANDYRL_GEISTER_TRAJ', '0') == '1'
    from handyrl_amd.actor_geister import GeisterMultiProcPool
    pool = GeisterMultiProcPool(args, n_games=cli.actors, seed=17,
                                workers=cli.workers, traj_mode=traj_mode,
                                make_stubs=False)

    use_cuda = torch.cuda.is_available()
    device = torch.device('cuda', 0) if use_cuda else torch.device('cpu')
    from handyrl_amd.envs.geister import Environment
    from handyrl_amd.train import Trainer
    torch.manual_seed(0)
    env = Environment()
    trainer = Trainer(args, env.net(), device=device, episodes=buffer,
                      batcher=batcher if batcher is not None else False)
    actor_model = env.net().to(device)
    actor_model.load_state_dict(trainer.model.state_dict())
    actor_model.eval()

    dreplay = dstep = None
    if cli.device_replay:
        from handyrl_amd.replay import TurnDeviceReplay
        from handyrl_amd.hipgraph import GraphedRecurrentTrainStep
        dreplay = TurnDeviceReplay(args, device, bytes_budget=2 << 30,
                                   ingest_thread=use_cuda)
    pool.attach(actor_model, device,
                replay=dreplay if traj_mode else None)

    def pump(n):
        frames = 0
        for _ in range(n * pool.calls_per_vec_step):
            frames += pool.step_once()
        eps = pool.harvest()
        if eps:
            (dreplay if dreplay is not None else buffer).extend(eps)
        return frames

    t0 = time.time()
    while pool.episodes_done < args['minimum_episodes']:
        pump(2)
    if dreplay is not None:
        dreplay.flush()
        dstep = GraphedRecurrentTrainStep(trainer, dreplay,
                                          args['batch_size'])
        print('# recurrent train-step captured: %s' % (dstep.graph is not None),
              file=sys.stderr, flush=True)
        n_prefill = len(dreplay)
    else:
        batcher.run()
        n_prefill = len(buffer)
    print('# prefill %d eps in %.1fs' % (n_prefill, time.time() - t0),
          file=sys.stderr, flush=True)

    def one_step():
        frames = pump(4)
        if dstep is not None:
            losses, dcnt = dstep.step()
        else:
            batch = batcher.batch()
            losses, dcnt = trainer.train_step(batch)
        # actors follow the trained weights (model push each step)
        actor_model.load_state_dict(trainer.model.state_dict())
        pool.refresh_weights()     # repack fused-DRC fragments
        return frames

    for _ in range(cli.warmup):
        one_step()
    if use_cuda:
        torch.cuda.synchronize()
    start = time.time()
    total = 0
    for _ in range(cli.steps):
        total += one_step()
    if use_cuda:
        torch.cuda.synchronize()
    el = time.time() - start
    pool.shutdown()

    print(json.dumps({
        'metric': 'geister_selfplay_env_frames_per_sec',
        'value': round(total / el, 1),
        'unit': 'frames/s', 'n_gpus': 1, 'steps': cli.steps,
        'warmup': cli.warmup, 'ms_per_step': round(1000 * el / cli.steps, 2),
        'higher_is_better': True, 'scaling': 'weak', 'vs_baseline': None,
        'dtype': 'fp32', 'data': 'synthetic',
        'config': {'model': 'GeisterNet(DRC 3x3 ConvLSTM)',
                   'actors': cli.actors, 'workers': cli.workers,
                   'global_batch': cli.batch_size, 'seq_len': 16,
                   'parallelism': 'dp1',
                   'learner_steps_per_sec': round(cli.steps / el, 2),
                   'loss': 'UPGO policy + TD(lambda) value/return'},
    }), flush=True)


if __name__ == '__main__':
    main()
