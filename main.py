"""handyrl_amd CLI — mode dispatch over config.yaml.

Modes (parity with the reference main.py):
  --train / -t            standalone training (learner + local workers)
  --train-server / -ts    training server for remote workers
  --worker / -w           remote worker cluster (connects to a server)
  --eval / -e             offline evaluation        [model(s)] [games] [procs]
  --eval-server / -es     network battle server     [games] [procs]
  --eval-client / -ec     network battle client     [model] [host]
"""

import sys

import yaml


if __name__ == '__main__':
    with open('config.yaml') as f:
        args = yaml.safe_load(f)
    print(args)

    if len(sys.argv) < 2:
        print('Please set a mode (see main.py docstring).')
        sys.exit(1)

    mode = sys.argv[1]

    if mode in ('--train', '-t'):
        from handyrl_amd.train import train_main as main
        main(args)
    elif mode in ('--train-server', '-ts'):
        from handyrl_amd.train import train_server_main as main
        main(args)
    elif mode in ('--worker', '-w'):
        from handyrl_amd.worker import worker_main as main
        main(args, sys.argv[2:])
    elif mode in ('--eval', '-e'):
        from handyrl_amd.evaluation import eval_main as main
        main(args, sys.argv[2:])
    elif mode in ('--eval-server', '-es'):
        from handyrl_amd.evaluation import eval_server_main as main
        main(args, sys.argv[2:])
    elif mode in ('--eval-client', '-ec'):
        from handyrl_amd.evaluation import eval_client_main as main
        main(args, sys.argv[2:])
    else:
        print('Not found mode %s.' % mode)
