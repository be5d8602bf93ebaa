"""Stochastic Weight Averaging over saved epoch checkpoints.

Usage: python scripts/aux_swa.py <env_name> [first_epoch] [last_epoch]
Averages models/{epoch}.pth state dicts with equal weight into
models/swa.pth (parity: reference scripts/aux_swa.py).
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from torch.optim.swa_utils import AveragedModel

from handyrl_amd.environment import make_env, prepare_env


def main():
    env_name = sys.argv[1] if len(sys.argv) > 1 else 'TicTacToe'
    first = int(sys.argv[2]) if len(sys.argv) > 2 else 1
    last = int(sys.argv[3]) if len(sys.argv) > 3 else None

    env_args = {'env': env_name}
    prepare_env(env_args)
    env = make_env(env_args)
    base = env.net()

    swa = AveragedModel(base)
    epoch = first
    used = 0
    while True:
        path = os.path.join('models', '%d.pth' % epoch)
        if not os.path.exists(path) or (last is not None and epoch > last):
            break
        base.load_state_dict(torch.load(path, map_location='cpu'))
        swa.update_parameters(base)
        used += 1
        epoch += 1

    if used == 0:
        print('no checkpoints found under models/')
        return
    out = os.path.join('models', 'swa.pth')
    torch.save(swa.module.state_dict(), out)
    print('averaged %d checkpoints -> %s' % (used, out))


if __name__ == '__main__':
    main()
