# Does the full pipeline LEARN? Track mean episode length + losses while
# training Hungry Geese self-play for N steps on one GPU.
import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from handyrl_amd.actor import MultiProcGeesePool
from handyrl_amd.models.geese_net import GeeseNet
from handyrl_amd.train import Trainer
from handyrl_amd.replay import DeviceReplay
from handyrl_amd.hipgraph import GraphedReplayTrainStep
from bench import bench_args

torch.set_num_threads(1)
args = bench_args(128, 16)
args['entropy_regularization'] = float(
    os.environ.get('HANDYRL_ENT_REG', args['entropy_regularization']))
traj = os.environ.get('HANDYRL_TRAJ', '1') == '1'
mpool = MultiProcGeesePool(args, n_games=1024, seed=7, workers=4,
                           traj_mode=traj)
device = torch.device('cuda', 0)
torch.cuda.set_device(device)
torch.manual_seed(0)
trainer = Trainer(args, GeeseNet(), device=device, batcher=False)
# HANDYRL_INGEST_THREAD=0: synchronous main-stream ingest (race probe)
ingest = os.environ.get('HANDYRL_INGEST_THREAD', '1') == '1'
replay = DeviceReplay(args, device, bytes_budget=2 << 30,
                      ingest_thread=ingest)
mpool.attach(trainer.model, device, replay=replay)
print('# slots=%d sync=%s ingest_thread=%s register=%s traj=%s' % (
    mpool.slots, os.environ.get('HANDYRL_ACTOR_SYNC', '0'), ingest,
    getattr(mpool, '_use_registered', False), traj), flush=True)

VALIDATE = os.environ.get('HANDYRL_VALIDATE') == '1'
min_prob_seen = [1.0]
bad_eps = [0]


def _check_eps(eps):
    """Cheap host-side validation of harvested episodes (pre-ingest)."""
    import numpy as np
    for ep in eps:
        alive = ep['alive']
        pr, v = ep['prob'][alive], ep['value'][alive]
        if pr.size:
            m = float(pr.min())
            if m < min_prob_seen[0]:
                min_prob_seen[0] = m
        if (not np.isfinite(pr).all() or (pr.size and pr.min() <= 0)
                or not np.isfinite(v).all()):
            bad_eps[0] += 1


def pump(n):
    frames = 0
    for _ in range(n * mpool.calls_per_vec_step):
        frames += mpool.step_once()
    eps = mpool.harvest()
    lens = [ep['steps'] for ep in eps]
    if VALIDATE:
        _check_eps(eps)
    replay.extend(eps)
    return frames, lens

while mpool.episodes_done < args['minimum_episodes']:
    pump(8)
replay.flush()
step = GraphedReplayTrainStep(trainer, replay, args['batch_size'])
assert step.graph is not None

N = int(sys.argv[1]) if len(sys.argv) > 1 else 400
window_lens, t0 = [], time.time()
import math
for i in range(N):
    frames, lens = pump(16)
    window_lens += lens
    losses, dcnt = step.step()
    mpool.refresh_weights()
    if VALIDATE:
        p = next(trainer.model.parameters())
        pf = bool(torch.isfinite(p).all())
        lf = math.isfinite(float(losses['p'])) and \
            math.isfinite(float(losses['v']))
        if not pf or not lf or bad_eps[0]:
            print('  probe step %d: params_finite=%s loss_finite=%s '
                  'bad_eps=%d min_prob=%.3e'
                  % (i + 1, pf, lf, bad_eps[0], min_prob_seen[0]), flush=True)
            if bad_eps[0] > 500 or ((not pf) and i > 25):
                break
    if (i + 1) % 10 == 0 and not math.isfinite(float(losses['p'])):
        print('NONFINITE_LOSS at step %d' % (i + 1), flush=True)
        if os.environ.get('HANDYRL_VALIDATE') == '1':
            torch.cuda.synchronize()
            import numpy as np
            host = step._last_host_idx
            for k in ('pos0', 'start', 'length', 'seat'):
                dev = step.idx[k].cpu().numpy()
                same = np.array_equal(dev, host[k])
                print('  idx[%s] device==host: %s  dev[min,max]=[%d,%d]'
                      % (k, same, dev.min(), dev.max()), flush=True)
            with torch.no_grad():
                batch = replay.gather_batch(
                    step.idx['pos0'], step.idx['start'], step.idx['length'],
                    step.idx['seat'], step.idx['outcome'],
                    step.idx['inv_total'])
                tm = batch['turn_mask'].bool().squeeze(-1)
                pr = batch['selected_prob'].squeeze(-1)[tm]
                vv = batch['value'].squeeze(-1)[tm]
                print('  eager regather: prob[min,max]=[%g,%g] n_nonpos=%d '
                      'n_nonfinite=%d | value[min,max]=[%g,%g] nonfinite=%d'
                      % (pr.min(), pr.max(), (pr <= 0).sum(),
                         (~torch.isfinite(pr)).sum(), vv.min(), vv.max(),
                         (~torch.isfinite(vv)).sum()), flush=True)
                p = next(trainer.model.parameters())
                print('  model params finite: %s'
                      % bool(torch.isfinite(p).all()), flush=True)
        break
    if (i + 1) % 50 == 0:
        torch.cuda.synchronize()
        ml = sum(window_lens) / max(1, len(window_lens))
        print('step %4d | mean_ep_len %.2f | eps %d | p %.4f v %.4f '
              'ent %.3f | guard_fires %d | %.1fs'
              % (i + 1, ml, len(window_lens),
                 float(losses['p']) / max(1.0, float(dcnt)),
                 float(losses['v']) / max(1.0, float(dcnt)),
                 float(losses['ent']) / max(1.0, float(dcnt)),
                 int(trainer.guard_fires.item()),
                 time.time() - t0), flush=True)
        window_lens = []
mpool.shutdown()
print('GUARD_FIRES_TOTAL %d' % int(trainer.guard_fires.item()))
print('LEARNING_CHECK_DONE')
