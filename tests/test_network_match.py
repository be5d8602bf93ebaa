"""Network battle protocol tests: the NetworkAgent (server stub) <->
NetworkAgentClient (RPC loop) pair driving exec_network_match, and a full
remote-worker training run over localhost TCP."""

import multiprocessing as mp
import os
import socket
import subprocess
import sys
import textwrap
import threading

import pytest

from handyrl_amd.agent import RandomAgent
from handyrl_amd.envs import tictactoe, geister
from handyrl_amd.evaluation import (NetworkAgent, NetworkAgentClient,
                                    exec_network_match)

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.parametrize('env_mod', [tictactoe, geister])
def test_network_match_rpc_roundtrip(env_mod):
    """Full match through the update/action/observe/outcome RPC with one
    replica env per remote agent (partial info for Geister)."""
    master = env_mod.Environment()
    players = master.players()

    conns = {}
    threads = []
    for p in players:
        server_end, client_end = mp.Pipe(duplex=True)
        conns[p] = server_end
        client = NetworkAgentClient(RandomAgent(), env_mod.Environment(), client_end)
        t = threading.Thread(target=client.run, daemon=True)
        t.start()
        threads.append((t, client_end))

    agents = {p: NetworkAgent(conns[p]) for p in players}
    result = exec_network_match(master, agents)
    assert result is not None
    outcome = result['result']
    assert set(outcome.keys()) == set(players)
    assert abs(sum(outcome.values())) < 1e-6

    for p in players:
        conns[p].send(('quit', []))
    for t, _ in threads:
        t.join(timeout=5)


def _free_port():
    s = socket.socket()
    s.bind(('', 0))
    port = s.getsockname()[1]
    s.close()
    return port


def test_remote_worker_training_over_tcp():
    """--train-server + --worker over localhost: entry handshake, gather
    data connections, episodes/results flowing back, one epoch saved."""
    entry_port, worker_port = _free_port(), _free_port()
    env = dict(os.environ, HANDYRL_ENTRY_PORT=str(entry_port),
               HANDYRL_WORKER_PORT=str(worker_port), PYTHONPATH=REPO)

    server_script = textwrap.dedent('''
        import sys
        sys.path.insert(0, %r)
        from handyrl_amd.train import train_server_main
        args = {
            'env_args': {'env': 'TicTacToe'},
            'train_args': {
                'turn_based_training': True, 'observation': False,
                'gamma': 0.8, 'forward_steps': 4, 'burn_in_steps': 0,
                'compress_steps': 4, 'entropy_regularization': 0.1,
                'entropy_regularization_decay': 0.1, 'update_episodes': 15,
                'batch_size': 4, 'minimum_episodes': 5, 'maximum_episodes': 200,
                'epochs': 1, 'num_batchers': 1, 'eval_rate': 0.1,
                'worker': {}, 'lambda': 0.7,
                'policy_target': 'TD', 'value_target': 'TD',
                'eval': {'opponent': ['random']}, 'seed': 0, 'restart_epoch': 0,
                'bf16': False,
            },
        }
        train_server_main(args)
        print('SERVER_DONE')
    ''') % REPO

    worker_script = textwrap.dedent('''
        import sys
        sys.path.insert(0, %r)
        from handyrl_amd.worker import worker_main
        args = {'worker_args': {'server_address': '127.0.0.1',
                                'num_parallel': 2, 'seed': 1}}
        worker_main(args, [])
    ''') % REPO

    import tempfile
    with tempfile.TemporaryDirectory() as tmp:
        server = subprocess.Popen([sys.executable, '-c', server_script],
                                  cwd=tmp, env=env, stdout=subprocess.PIPE,
                                  stderr=subprocess.STDOUT, text=True)
        import time
        time.sleep(3)                       # let the entry server bind
        worker = subprocess.Popen([sys.executable, '-c', worker_script],
                                  cwd=tmp, env=env, stdout=subprocess.PIPE,
                                  stderr=subprocess.STDOUT, text=True)
        try:
            out, _ = server.communicate(timeout=240)
        finally:
            worker.kill()
            worker.communicate(timeout=30)
        assert 'SERVER_DONE' in out, out[-3000:]
        assert 'updated model(' in out
        assert 'accepted connection' in out
        assert os.path.exists(os.path.join(tmp, 'models', '1.pth'))
