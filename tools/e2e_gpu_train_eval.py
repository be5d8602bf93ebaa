"""End-to-end GPU run of the CONFIG-DRIVEN stack: main.py-style training
(Learner + GPU actor pool + device replay + CPU eval workers) for a few
epochs, then the offline evaluation harness on the saved checkpoint vs
random opponents.

Exercises in one go: Learner orchestration with the multiprocess
HungryGeese pool (traj mode), epoch rollover + reference-layout
checkpoint save, the in-training eval role split, and `--eval`-style
evaluate_mp on models/latest.pth.  Run from a scratch directory.
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    epochs = int(sys.argv[1]) if len(sys.argv) > 1 else 3
    env = sys.argv[2] if len(sys.argv) > 2 else 'HungryGeese'
    geese = env == 'HungryGeese'
    args = {
        'env_args': {'env': env},
        'train_args': {
            'turn_based_training': not geese, 'observation': False,
            'gamma': 0.8, 'forward_steps': 16, 'burn_in_steps': 0,
            'compress_steps': 4, 'entropy_regularization': 0.1,
            'entropy_regularization_decay': 0.1,
            'update_episodes': 20000 if geese else 6000,
            'batch_size': 128 if geese else 32,
            'minimum_episodes': 2000 if geese else 400,
            'maximum_episodes': 40000,
            'epochs': epochs, 'num_batchers': 2, 'eval_rate': 0.1,
            'worker': {'type': 'gpu',
                       'num_envs': 2048,
                       'num_actor_procs': 8, 'num_parallel': 2},
            'replay': 'device',
            'lambda': 0.7,
            'policy_target': 'VTRACE' if geese else 'UPGO',
            'value_target': 'VTRACE' if geese else 'TD',
            'eval': {'opponent': ['random']},
            'seed': 0, 'restart_epoch': 0, 'bf16': geese,
            'compress_episodes': False,
        },
    }
    from handyrl_amd.train import train_main
    train_main(args)
    print('TRAIN_DONE', flush=True)

    # offline eval in a FRESH interpreter: forking eval workers from a
    # CUDA-initialized process deadlocks in HIP
    import subprocess
    code = (
        "import sys; sys.path.insert(0, %r);"
        "from handyrl_amd.evaluation import eval_main;"
        "eval_main({'env_args': {'env': %r}},"
        "          ['models/latest.pth', '64', '4'])"
        % (os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
           env))
    proc = subprocess.run([sys.executable, '-c', code], timeout=600)
    assert proc.returncode == 0, proc.returncode
    print('EVAL_DONE', flush=True)
    # skip interpreter teardown: daemon threads holding HIP state abort
    # noisily inside libc at exit (cosmetic, but it flips the exit code)
    os._exit(0)


if __name__ == '__main__':
    main()
