import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
# Time train-step components: batch H2D copy, graph replay, weight refresh.
import time, torch
from handyrl_amd.actor import GeeseActorPool
from handyrl_amd.batch import make_batch
from handyrl_amd.models.geese_net import GeeseNet
from handyrl_amd.train import Trainer
from handyrl_amd.util import map_r
from bench import bench_args

args = bench_args(128, 16)
dev = torch.device('cuda', 0)
torch.manual_seed(0)
trainer = Trainer(args, GeeseNet(), device=dev)
pool = GeeseActorPool(trainer.model, args, n_games=256, device=dev, seed=0)
trainer.model.eval()
while pool.episodes_done < 170:
    pool.step_once()
trainer.episodes.extend(pool.harvest())
batch = make_batch([trainer.episodes.select_episode() for _ in range(128)], args)
sz = sum(map(lambda t: t.numel()*t.element_size(), batch.values() if isinstance(batch, dict) else []))
flat = []
map_r(batch, flat.append)
print('batch bytes = %.1f MB' % (sum(t.numel()*t.element_size() for t in flat)/1e6))

trainer.enable_cuda_graph(batch)
assert trainer.graphed_step is not None
g = trainer.graphed_step

def timeit(label, fn, n=30):
    torch.cuda.synchronize(); t=time.time()
    for _ in range(n): fn()
    torch.cuda.synchronize()
    print('%s: %.2f ms' % (label, 1000*(time.time()-t)/n))

from handyrl_amd.hipgraph import _copy_into
timeit('copy batch->static (pageable)', lambda: _copy_into(g.static, batch))
pinned = map_r(batch, lambda t: t.pin_memory())
timeit('copy pinned->static', lambda: _copy_into(g.static, pinned))
timeit('replay', lambda: g.graph.replay())
timeit('refresh_weights (eager)', lambda: pool.fused.refresh())
timeit('full graphed train_step', lambda: trainer.train_step(batch))
# actor step components under load
timeit('actor step_once', lambda: pool.step_once(), n=50)

from torch.profiler import profile, ProfilerActivity
with profile(activities=[ProfilerActivity.CUDA]) as prof:
    for _ in range(10):
        g.graph.replay()
    torch.cuda.synchronize()
ka = prof.key_averages()
tot = sum(e.device_time_total for e in ka)
print('TRAIN REPLAY: total device time per replay: %.2f ms' % (tot/1e3/10))
for e in sorted(ka, key=lambda e: -e.device_time_total)[:14]:
    print('%9.1f us  x%-5d %s' % (e.device_time_total/10, e.count//10, e.key[:90]))
