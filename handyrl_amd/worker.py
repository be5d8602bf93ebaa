"""Worker tree: leaf actors, gather relay nodes, and the local/remote
cluster frontends.

Topology and protocol contract (reference worker.py): learner <-> gathers
(one per ~16 workers) <-> workers; job args are requested with
``('args', None)``, models pulled by id with ``('model', id)``, episodes
and results pushed as ``('episode'/'result', payload)``.  The gather tier
batches args requests upstream, caches model responses by id and buffers
uploads.  Remote mode keeps the entry handshake on :9999 (assigning
``base_worker_id`` ranges) and gather data connections on :9998; workers
may join or leave at any time.

The implementation is this repo's own: the relay's three concerns live in
small dedicated components, and the leaf worker separates job transport
from job execution.
"""

import copy
import pickle
import random
import threading
import time
import multiprocessing as mp
from socket import gethostname

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


def _gathers_for(num_parallel):
    """Gather processes for a worker count (~16 leaves per relay)."""
    return 1 + max(0, num_parallel - 1) // 16


class _LeafModelPool:
    """Model cache on a leaf worker: capacity one (the live epoch), plus
    the probe-built RandomModel for id 0."""

    def __init__(self, env, fetch):
        self.env = env
        self.fetch = fetch             # model_id -> pickled nn.Module bytes
        self.entries = {}

    def get(self, model_id):
        return self.entries.get(model_id)

    def ensure(self, model_ids):
        for model_id in model_ids:
            if model_id is None or model_id < 0 or model_id in self.entries:
                continue
            model = pickle.loads(self.fetch(model_id))
            if model_id == 0:
                # epoch-0 opponent: uniform-random model probed on a real obs
                self.env.reset()
                probe_obs = self.env.observation(self.env.players()[0])
                model = RandomModel(model, probe_obs)
            while len(self.entries) >= 1:
                self.entries.pop(next(iter(self.entries)))
            self.entries[model_id] = ModelWrapper(model)


class Worker:
    """Leaf actor: pulls job args, runs generation or evaluation, pushes
    the episode/result back."""

    def __init__(self, args, conn, wid):
        print('opened worker %d' % wid)
        self.worker_id = wid
        self.args = args
        self.conn = conn
        self.env = make_env({**args['env'], 'id': wid})
        self.generator = Generator(self.env, args)
        self.evaluator = Evaluator(self.env, args)
        self.models = _LeafModelPool(
            self.env, lambda mid: send_recv(self.conn, ('model', mid)))
        random.seed(args['seed'] + wid)

    def __del__(self):
        print('closed worker %d' % self.worker_id)

    def _job_models(self, job):
        if 'model_id' not in job:
            return {}
        self.models.ensure(job['model_id'].values())
        return {p: self.models.get(mid)
                for p, mid in job['model_id'].items()}

    def run(self):
        while True:
            job = send_recv(self.conn, ('args', None))
            if job is None:
                break
            models = self._job_models(job)
            if job['role'] == 'g':
                payload = ('episode', self.generator.execute(models, job))
            elif job['role'] == 'e':
                payload = ('result', self.evaluator.execute(models, job))
            else:
                continue
            send_recv(self.conn, payload)


def make_worker_args(args, n_ga, gaid, base_wid, wid, conn):
    # worker ids interleave across gathers: gather g's k-th worker gets
    # base + k*n_gathers + g, so ids are globally distinct and dense
    return args, conn, base_wid + wid * n_ga + gaid


def open_worker(args, conn, wid):
    Worker(args, conn, wid).run()


# -- gather relay ------------------------------------------------------------

class _ArgsPrefetcher:
    """Pulls job-args from upstream in bursts and hands them out one by
    one (cuts per-job round trips by the burst factor)."""

    def __init__(self, upstream, burst):
        self.upstream = upstream
        self.burst = burst
        self.pending = []

    def next(self):
        if not self.pending:
            self.upstream.send(('args', [None] * self.burst))
            self.pending = list(self.upstream.recv())
        return self.pending.pop(0)


class _UploadBuffer:
    """Accumulates episode/result uploads and flushes them upstream in
    batches (one round trip per burst, per command kind)."""

    def __init__(self, upstream, burst):
        self.upstream = upstream
        self.burst = burst
        self.held = {}
        self.count = 0

    def add(self, command, payload):
        self.held.setdefault(command, []).append(payload)
        self.count += 1
        if self.count >= self.burst:
            self.flush()

    def flush(self):
        for command, payloads in self.held.items():
            self.upstream.send((command, payloads))
            self.upstream.recv()
        self.held = {}
        self.count = 0


class Gather(QueueCommunicator):
    """Relay node between the learner and ~16 leaf workers: prefetches job
    args, caches model pulls by id, batches uploads."""

    def __init__(self, args, conn, gaid):
        print('started gather %d' % gaid)
        super().__init__()
        self.gather_id = gaid
        self.server_conn = conn

        wcfg = args['worker']
        n_pro, n_ga = wcfg['num_parallel'], wcfg['num_gathers']
        local_workers = n_pro // n_ga + (1 if gaid < n_pro % n_ga else 0)
        base_wid = wcfg.get('base_worker_id', 0)
        for leaf in open_multiprocessing_connections(
                local_workers, open_worker,
                lambda wid, c: make_worker_args(args, n_ga, gaid, base_wid,
                                                wid, c)):
            self.add_connection(leaf)

        burst = 1 + local_workers // 4
        self.args_feed = _ArgsPrefetcher(conn, burst)
        self.uploads = _UploadBuffer(conn, burst)
        self.model_cache = {}

    def __del__(self):
        print('finished gather %d' % self.gather_id)

    def _pull_model(self, model_id):
        if model_id not in self.model_cache:
            self.server_conn.send(('model', model_id))
            self.model_cache[model_id] = self.server_conn.recv()
        return self.model_cache[model_id]

    def run(self):
        import queue as _queue
        while self.connection_count() > 0:
            try:
                conn, (command, payload) = self.recv(timeout=0.3)
            except _queue.Empty:
                continue
            if command == 'args':
                self.send(conn, self.args_feed.next())
            elif command == 'model':
                self.send(conn, self._pull_model(payload))
            else:                       # episode / result upload
                self.send(conn, None)   # ack immediately, flush in bursts
                self.uploads.add(command, payload)


def gather_loop(args, conn, gaid):
    Gather(args, conn, gaid).run()


# -- cluster frontends -------------------------------------------------------

class WorkerCluster(QueueCommunicator):
    """Local mode: gather+worker tree as child processes over pipes."""

    def __init__(self, args):
        super().__init__()
        self.args = args

    def run(self):
        wcfg = self.args['worker']
        if 'num_gathers' not in wcfg:
            wcfg['num_gathers'] = _gathers_for(wcfg['num_parallel'])
        for gaid in range(wcfg['num_gathers']):
            here, there = mp.Pipe(duplex=True)
            mp.Process(target=gather_loop,
                       args=(self.args, there, gaid)).start()
            there.close()
            self.add_connection(here)


class WorkerServer(QueueCommunicator):
    """Remote mode server.

    Entry thread (:9999): each connecting machine sends its worker args,
    receives the full config back with a disjoint ``base_worker_id``
    range.  Data thread (:9998): accepts gather connections into the hub.
    """

    def __init__(self, args):
        super().__init__()
        self.args = args
        self.total_worker_count = 0

    def _entry_loop(self, port):
        print('started entry server %d' % port)
        for conn in accept_socket_connections(port=port):
            worker_args = conn.recv()
            print('accepted connection from %s!' % worker_args['address'])
            worker_args['base_worker_id'] = self.total_worker_count
            self.total_worker_count += worker_args['num_parallel']
            reply = copy.deepcopy(self.args)
            reply['worker'] = worker_args
            conn.send(reply)
            conn.close()

    def _data_loop(self, port):
        print('started worker server %d' % port)
        for conn in accept_socket_connections(port=port):
            self.add_connection(conn)

    def run(self):
        threading.Thread(target=self._entry_loop, args=(ENTRY_PORT,),
                         daemon=True).start()
        threading.Thread(target=self._data_loop, args=(WORKER_PORT,),
                         daemon=True).start()


def entry(worker_args):
    conn = connect_socket_connection(worker_args['server_address'],
                                     ENTRY_PORT)
    full_args = send_recv(conn, worker_args)
    conn.close()
    return full_args


class RemoteWorkerCluster:
    """Remote machine side: entry handshake, then one gather process per
    data connection back to the learner."""

    def __init__(self, args):
        args['address'] = gethostname()
        if 'num_gathers' not in args:
            args['num_gathers'] = _gathers_for(args['num_parallel'])
        self.args = args

    def run(self):
        full_args = entry(self.args)
        print(full_args)
        prepare_env(full_args['env'])
        children = []
        try:
            for gaid in range(self.args['num_gathers']):
                conn = connect_socket_connection(
                    self.args['server_address'], WORKER_PORT)
                proc = mp.Process(target=gather_loop,
                                  args=(full_args, conn, gaid))
                proc.start()
                conn.close()
                children.append(proc)
            while True:
                time.sleep(100)
        finally:
            for proc in children:
                proc.terminate()


def worker_main(args, argv):
    worker_args = args['worker_args']
    if len(argv) >= 1:
        worker_args['num_parallel'] = int(argv[0])
    RemoteWorkerCluster(args=worker_args).run()
