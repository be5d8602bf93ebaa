"""Worker tree: leaf actors, gather aggregation nodes, and the local/remote
cluster frontends (parity: reference worker.py).

Topology: learner <-> gathers (one per ~16 workers) <-> workers, with
request batching, model-response caching and upload buffering in the
gather tier.  Remote mode: entry handshake on :9999, gather data
connections on :9998; workers may join or leave at any time.
"""

import copy
import functools
import pickle
import queue
import random
import threading
import time
import multiprocessing as mp
from socket import gethostname
from collections import deque

from .environment import prepare_env, make_env
from .connection import QueueCommunicator
from .connection import send_recv, open_multiprocessing_connections
from .connection import connect_socket_connection, accept_socket_connections
from .evaluation import Evaluator
from .generation import Generator
from .model import ModelWrapper, RandomModel

import os as _os
ENTRY_PORT = int(_os.environ.get('HANDYRL_ENTRY_PORT', 9999))
WORKER_PORT = int(_os.environ.get('HANDYRL_WORKER_PORT', 9998))


class Worker:
    """Leaf actor: pulls job args, runs generation or evaluation, pushes
    the episode/result back."""

    def __init__(self, args, conn, wid):
        print('opened worker %d' % wid)
        self.worker_id = wid
        self.args = args
        self.conn = conn
        self.model_pool = {}

        self.env = make_env({**args['env'], 'id': wid})
        self.generator = Generator(self.env, self.args)
        self.evaluator = Evaluator(self.env, self.args)

        random.seed(args['seed'] + wid)

    def __del__(self):
        print('closed worker %d' % self.worker_id)

    def _fetch_models(self, model_ids):
        for model_id in model_ids:
            if model_id is None or model_id < 0 or model_id in self.model_pool:
                continue
            model = pickle.loads(send_recv(self.conn, ('model', model_id)))
            if model_id == 0:
                # epoch 0 opponent: uniform-random model probed on a real obs
                self.env.reset()
                obs = self.env.observation(self.env.players()[0])
                model = RandomModel(model, obs)
            if len(self.model_pool) >= 1:      # keep a pool of one
                self.model_pool.pop(next(iter(self.model_pool)))
            self.model_pool[model_id] = ModelWrapper(model)

    def run(self):
        while True:
            args = send_recv(self.conn, ('args', None))
            if args is None:
                break
            role = args['role']

            models = {}
            if 'model_id' in args:
                self._fetch_models(list(args['model_id'].values()))
                for p, model_id in args['model_id'].items():
                    models[p] = self.model_pool.get(model_id, None)

            if role == 'g':
                episode = self.generator.execute(models, args)
                send_recv(self.conn, ('episode', episode))
            elif role == 'e':
                result = self.evaluator.execute(models, args)
                send_recv(self.conn, ('result', result))


def make_worker_args(args, n_ga, gaid, base_wid, wid, conn):
    return args, conn, base_wid + wid * n_ga + gaid


def open_worker(args, conn, wid):
    worker = Worker(args, conn, wid)
    worker.run()


class Gather(QueueCommunicator):
    """Aggregation node: batches args requests upstream, caches model
    responses by id, and buffers episode/result uploads."""

    def __init__(self, args, conn, gaid):
        print('started gather %d' % gaid)
        super().__init__()
        self.gather_id = gaid
        self.server_conn = conn
        self.args_queue = deque()
        self.data_map = {'model': {}}
        self.result_send_map = {}
        self.result_send_cnt = 0

        n_pro, n_ga = args['worker']['num_parallel'], args['worker']['num_gathers']
        num_workers_here = (n_pro // n_ga) + int(gaid < n_pro % n_ga)
        base_wid = args['worker'].get('base_worker_id', 0)

        worker_conns = open_multiprocessing_connections(
            num_workers_here, open_worker,
            functools.partial(make_worker_args, args, n_ga, gaid, base_wid))
        for conn_ in worker_conns:
            self.add_connection(conn_)

        self.buffer_length = 1 + len(worker_conns) // 4

    def __del__(self):
        print('finished gather %d' % self.gather_id)

    def run(self):
        while self.connection_count() > 0:
            try:
                conn, (command, args) = self.recv(timeout=0.3)
            except queue.Empty:
                continue

            if command == 'args':
                if len(self.args_queue) == 0:
                    self.server_conn.send((command, [None] * self.buffer_length))
                    self.args_queue += self.server_conn.recv()
                self.send(conn, self.args_queue.popleft())

            elif command in self.data_map:
                data_id = args
                if data_id not in self.data_map[command]:
                    self.server_conn.send((command, args))
                    self.data_map[command][data_id] = self.server_conn.recv()
                self.send(conn, self.data_map[command][data_id])

            else:
                # ack first, upload in buffered bursts
                self.send(conn, None)
                self.result_send_map.setdefault(command, []).append(args)
                self.result_send_cnt += 1
                if self.result_send_cnt >= self.buffer_length:
                    for cmd, args_list in self.result_send_map.items():
                        self.server_conn.send((cmd, args_list))
                        self.server_conn.recv()
                    self.result_send_map = {}
                    self.result_send_cnt = 0


def gather_loop(args, conn, gaid):
    gather = Gather(args, conn, gaid)
    gather.run()


class WorkerCluster(QueueCommunicator):
    """Local mode: gathers+workers as child processes over pipes."""

    def __init__(self, args):
        super().__init__()
        self.args = args

    def run(self):
        if 'num_gathers' not in self.args['worker']:
            self.args['worker']['num_gathers'] = \
                1 + max(0, self.args['worker']['num_parallel'] - 1) // 16
        for i in range(self.args['worker']['num_gathers']):
            conn0, conn1 = mp.Pipe(duplex=True)
            mp.Process(target=gather_loop, args=(self.args, conn1, i)).start()
            conn1.close()
            self.add_connection(conn0)


class WorkerServer(QueueCommunicator):
    """Remote mode server: entry handshake (:9999) assigns worker id ranges
    and returns the full config; gather data connections accepted on :9998."""

    def __init__(self, args):
        super().__init__()
        self.args = args
        self.total_worker_count = 0

    def run(self):
        def entry_server(port):
            print('started entry server %d' % port)
            acceptor = accept_socket_connections(port=port)
            while True:
                conn = next(acceptor)
                worker_args = conn.recv()
                print('accepted connection from %s!' % worker_args['address'])
                worker_args['base_worker_id'] = self.total_worker_count
                self.total_worker_count += worker_args['num_parallel']
                args = copy.deepcopy(self.args)
                args['worker'] = worker_args
                conn.send(args)
                conn.close()

        def worker_server(port):
            print('started worker server %d' % port)
            acceptor = accept_socket_connections(port=port)
            while True:
                self.add_connection(next(acceptor))

        threading.Thread(target=entry_server, args=(ENTRY_PORT,), daemon=True).start()
        threading.Thread(target=worker_server, args=(WORKER_PORT,), daemon=True).start()


def entry(worker_args):
    conn = connect_socket_connection(worker_args['server_address'], ENTRY_PORT)
    conn.send(worker_args)
    args = conn.recv()
    conn.close()
    return args


class RemoteWorkerCluster:
    """Remote machine side: handshake, then one gather process per data
    connection to the learner."""

    def __init__(self, args):
        args['address'] = gethostname()
        if 'num_gathers' not in args:
            args['num_gathers'] = 1 + max(0, args['num_parallel'] - 1) // 16
        self.args = args

    def run(self):
        args = entry(self.args)
        print(args)
        prepare_env(args['env'])

        processes = []
        try:
            for i in range(self.args['num_gathers']):
                conn = connect_socket_connection(self.args['server_address'], WORKER_PORT)
                p = mp.Process(target=gather_loop, args=(args, conn, i))
                p.start()
                conn.close()
                processes.append(p)
            while True:
                time.sleep(100)
        finally:
            for p in processes:
                p.terminate()


def worker_main(args, argv):
    worker_args = args['worker_args']
    if len(argv) >= 1:
        worker_args['num_parallel'] = int(argv[0])
    RemoteWorkerCluster(args=worker_args).run()
