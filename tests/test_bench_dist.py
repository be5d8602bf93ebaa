"""bench.py's distributed path on CPU: 2 gloo ranks through the real
entrypoint (the same code the driver's multi-GPU SCALE run launches with
torch.distributed.run over RCCL), small sizes."""

import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_two_rank_gloo():
    env = dict(os.environ)
    env.pop('RANK', None)
    env.pop('WORLD_SIZE', None)
    port = str(20000 + os.getpid() % 20000)
    res = subprocess.run(
        [sys.executable, '-m', 'torch.distributed.run', '--nnodes=1',
         '--nproc-per-node', '2', '--master-addr', '127.0.0.1',
         '--master-port', port, 'bench.py', '--gpus', '2',
         '--steps', '2', '--warmup', '1', '--envs', '48',
         '--batch-size', '4', '--forward-steps', '8'],
        cwd=REPO, env=env, capture_output=True, text=True, timeout=420)
    out = res.stdout
    assert res.returncode == 0, (out[-3000:], res.stderr[-3000:])
    # exactly one JSON line, from rank 0, with the whole-job aggregate
    import json
    lines = [l for l in out.splitlines() if l.startswith('{')]
    assert len(lines) == 1, out[-2000:]
    rec = json.loads(lines[0])
    assert rec['n_gpus'] == 2
    assert rec['value'] > 0
    assert rec['config']['parallelism'] == 'dp2'
