# Reproducer/detector for the gated actor-transport corruption
# (BASELINE.md "learning sanity at round-1 close").  Pumps the
# multi-process actor pool and scans every harvested episode for
# corrupt recorded outputs: non-finite values, probs outside (0, 1],
# or actions outside [0, 4).  Healthy runs report zero findings.
#
# Usage on a GPU box:
#   HANDYRL_ACTOR_SLOTS=2 python tools/repro_transport_race.py   # race on
#   HANDYRL_ACTOR_SLOTS=1 python tools/repro_transport_race.py   # control
#   HANDYRL_ACTOR_SLOTS=2 HANDYRL_ACTOR_SYNC=1 python ...        # no-pipeline probe
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, torch
torch.set_num_threads(1)
from handyrl_amd.actor import MultiProcGeesePool
from handyrl_amd.models.geese_net import GeeseNet
from bench import bench_args

args = bench_args(128, 16)
mpool = MultiProcGeesePool(args, n_games=1024, seed=7, workers=4)
device = torch.device('cuda', 0)
torch.cuda.set_device(device)
torch.manual_seed(0)
model = GeeseNet().to(device)
model.eval()
mpool.attach(model, device)
print('# slots=%d sync=%s register=%s' % (
    mpool.slots, os.environ.get('HANDYRL_ACTOR_SYNC', '0'),
    getattr(mpool, '_use_registered', False)), flush=True)

bad = 0
eps_seen = 0
t0 = time.time()
for it in range(200):
    for _ in range(mpool.calls_per_vec_step * 4):
        mpool.step_once()
    for ep in mpool.harvest():
        eps_seen += 1
        alive = ep['alive']
        v, pr, ac = ep['value'], ep['prob'], ep['action']
        probs_live = pr[alive]
        errs = []
        if not np.isfinite(v).all():
            errs.append('nonfinite value')
        if not np.isfinite(pr).all() or (probs_live <= 0).any() or (probs_live > 1.0001).any():
            errs.append('bad prob')
        if (ac[alive] < 0).any() or (ac[alive] >= 4).any():
            errs.append('bad action')
        if errs:
            bad += 1
            if bad <= 5:
                print('CORRUPT ep (steps=%d): %s | v range [%g, %g] | prob range [%g, %g]'
                      % (ep['steps'], ','.join(errs), np.nanmin(v), np.nanmax(v),
                         np.nanmin(pr), np.nanmax(pr)), flush=True)
    if bad and it > 20:
        break
print('RESULT: %d corrupt / %d episodes in %.1fs' % (bad, eps_seen, time.time() - t0))
mpool.shutdown()
