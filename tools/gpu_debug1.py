import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import sys, time, torch
mode = sys.argv[1]
from handyrl_amd.actor import GeeseActorPool
from handyrl_amd.batch import make_batch, EpisodeBuffer
from handyrl_amd.models.geese_net import GeeseNet
from handyrl_amd.train import Trainer
from bench import bench_args

args = bench_args(128, 16)
device = torch.device('cuda', 0)
torch.manual_seed(1234)
trainer = Trainer(args, GeeseNet(), device=device)
pool = GeeseActorPool(trainer.model, args, n_games=256, device=device, seed=1000)
trainer.model.eval()

if mode == 'actor':
    for i in range(400):
        pool.step_once()
        if i % 50 == 0:
            torch.cuda.synchronize(); print('actor step', i, 'eps', pool.episodes_done, flush=True)
    torch.cuda.synchronize()
    print('ACTOR_OK', pool.episodes_done, flush=True)
elif mode == 'learner':
    while pool.episodes_done < 170:
        pool.step_once()
    trainer.episodes.extend(pool.harvest())
    print('buffer', len(trainer.episodes), flush=True)
    trainer.model.train()
    for i in range(10):
        sel = [trainer.episodes.select_episode() for _ in range(args['batch_size'])]
        batch = make_batch(sel, args)
        losses, dcnt = trainer.train_step(batch)
        torch.cuda.synchronize()
        print('learner step', i, 'loss', float(losses['total']), 'dcnt', dcnt, flush=True)
    print('LEARNER_OK', flush=True)
