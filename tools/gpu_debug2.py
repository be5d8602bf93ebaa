import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
# Bisect which train-step stage breaks hipGraph capture.
import torch, torch.nn as nn
from handyrl_amd.models.geese_net import GeeseNet
from handyrl_amd.train import compute_loss, Trainer
from handyrl_amd.actor import GeeseActorPool
from handyrl_amd.batch import make_batch
from handyrl_amd.util import map_r
from bench import bench_args

args = bench_args(32, 16)
dev = torch.device('cuda', 0)
trainer = Trainer(args, GeeseNet(), device=dev)
pool = GeeseActorPool(trainer.model, args, n_games=64, device=dev, seed=0, use_graphs=False)
trainer.model.eval()
while pool.episodes_done < 40:
    pool.step_once()
trainer.episodes.extend(pool.harvest())
sel = [trainer.episodes.select_episode() for _ in range(args['batch_size'])]
batch = make_batch(sel, args)
static = map_r(batch, lambda t: t.to(dev).clone())
trainer.model.train()
for g in trainer.optimizer.param_groups: g['capturable'] = True

def stage(n):
    with torch.autocast('cuda', dtype=torch.bfloat16):
        losses, dcnt = compute_loss(static, trainer.wrapped_model, None, args)
    if n >= 2:
        trainer.optimizer.zero_grad(set_to_none=False)
        losses['total'].backward()
    if n >= 3:
        nn.utils.clip_grad_norm_(trainer.params, 4.0)
    if n >= 4:
        trainer.optimizer.step()
    return losses

for n in (1, 2, 3, 4):
    try:
        s = torch.cuda.Stream(); s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2): stage(n)
        torch.cuda.current_stream().wait_stream(s)
        gr = torch.cuda.CUDAGraph()
        with torch.cuda.graph(gr):
            stage(n)
        gr.replay(); torch.cuda.synchronize()
        print('stage %d CAPTURE OK' % n, flush=True)
    except Exception as e:
        print('stage %d FAILED: %s' % (n, str(e).splitlines()[0]), flush=True)
