"""Host-side control-plane transport.

Carries job args, episodes and results (KB-scale messages) between the
learner, gather relays and workers.  The bandwidth-critical GPU paths
(gradient all-reduce, model broadcast) do NOT go through here — they use
RCCL over xGMI (handyrl_amd/dist.py).

Wire compatibility contract (so remote CPU workers running against a
reference-style deployment interoperate, reference connection.py:20-69):
every socket message is a 4-byte network-order signed length followed by a
pickle payload.  Local transport is ``multiprocessing.Pipe``; both carriers
expose the same ``send``/``recv``/``fileno`` surface, which is what lets
the whole distributed tree run single-machine.

Beyond the wire format, the implementations here are this repo's own:
buffered-file socket IO instead of manual chunk loops, an event-driven
single-thread job pool, and a hub with explicit per-connection accounting
and clean shutdown.
"""

import os
import pickle
import queue
import socket
import struct
import threading
import multiprocessing as mp
import multiprocessing.connection as mp_connection

_HEADER = struct.Struct('!i')
_COALESCE_LIMIT = 16384


def send_recv(conn, sdata):
    """One request/response round trip."""
    conn.send(sdata)
    return conn.recv()


class PickledConnection:
    """Length-framed pickle messages over a stream socket.

    Reads go through a buffered file object (the kernel-to-user copies are
    batched instead of looped 4-or-N bytes at a time); writes use
    ``sendall`` with small messages coalesced into a single syscall.
    """

    def __init__(self, conn):
        self.conn = conn
        self._rfile = conn.makefile('rb', buffering=1 << 16)

    def __del__(self):
        self.close()

    def close(self):
        if getattr(self, 'conn', None) is None:
            return
        try:
            self._rfile.close()
        except OSError:
            pass
        self.conn.close()
        self.conn = None

    def fileno(self):
        return self.conn.fileno()

    def _read_exact(self, n):
        data = self._rfile.read(n)
        if data is None or len(data) != n:
            raise ConnectionResetError('peer closed mid-frame')
        return data

    def recv(self):
        size, = _HEADER.unpack(self._read_exact(4))
        if size < 0:
            raise ConnectionResetError('negative frame length')
        return pickle.loads(self._read_exact(size)) if size else None

    def send(self, msg):
        body = pickle.dumps(msg)
        if 0 < len(body) <= _COALESCE_LIMIT:
            self.conn.sendall(_HEADER.pack(len(body)) + body)
        else:
            self.conn.sendall(_HEADER.pack(len(body)))
            if body:
                self.conn.sendall(body)


# -- socket plumbing ---------------------------------------------------------

def open_socket_connection(port, reuse=False):
    sock = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
    sock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
    sock.bind(('', int(port)))
    return sock


def accept_socket_connection(sock):
    try:
        conn, _addr = sock.accept()
        return PickledConnection(conn)
    except socket.timeout:
        return None


def listen_socket_connections(n, port):
    sock = open_socket_connection(port)
    sock.listen(n)
    return [accept_socket_connection(sock) for _ in range(n)]


def connect_socket_connection(host, port):
    sock = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
    try:
        sock.connect((host, int(port)))
    except ConnectionRefusedError:
        print('failed to connect %s %d' % (host, port))
    return PickledConnection(sock)


def accept_socket_connections(port, timeout=None, maxsize=1024):
    """Generator yielding accepted connections (None on accept timeout);
    peers may connect at any time for the lifetime of the acceptor."""
    sock = open_socket_connection(port)
    sock.listen(maxsize)
    sock.settimeout(timeout)
    accepted = 0
    while accepted < maxsize:
        conn = accept_socket_connection(sock)
        if conn is not None:
            accepted += 1
        yield conn


def open_multiprocessing_connections(num_process, target, args_func):
    """Spawn ``num_process`` children, each owning one end of a duplex pipe;
    returns the parent-side connections."""
    parents = []
    for i in range(num_process):
        here, there = mp.Pipe(duplex=True)
        mp.Process(target=target, args=args_func(i, there)).start()
        there.close()
        parents.append(here)
    return parents


# -- fan-out job pool --------------------------------------------------------

class MultiProcessJobExecutor:
    """Event-driven fan-out pool over ``num_workers`` pipe-connected child
    processes.

    A single dispatcher thread owns the whole lifecycle: it primes every
    worker with one job from ``send_generator``, then waits on the pipe set
    and refills each worker the moment its result arrives.  (One thread,
    no idle-worker queue: readiness IS the wait-set event.)  Results pass
    through ``postprocess`` and land in a bounded output queue that
    throttles the generator when the consumer falls behind.
    """

    def __init__(self, func, send_generator, num_workers, postprocess=None,
                 out_depth=8):
        self._jobs = send_generator
        self._post = postprocess
        self.output_queue = queue.Queue(maxsize=out_depth)
        self.shutdown_flag = False
        self.conns = []
        for i in range(num_workers):
            here, there = mp.Pipe(duplex=True)
            mp.Process(target=func, args=(there, i), daemon=True).start()
            there.close()
            self.conns.append(here)

    def recv(self):
        return self.output_queue.get()

    def start(self):
        threading.Thread(target=self._dispatch, daemon=True).start()

    def _dispatch(self):
        try:
            for conn in self.conns:
                conn.send(next(self._jobs))
            while not self.shutdown_flag:
                for conn in mp_connection.wait(self.conns, timeout=0.3):
                    result = conn.recv()
                    conn.send(next(self._jobs))  # refill before postprocess
                    if self._post is not None:
                        result = self._post(result)
                    self.output_queue.put(result)
        except (OSError, EOFError, BrokenPipeError, StopIteration):
            return      # workers gone / generator closed: quiet shutdown
        except Exception:   # noqa: BLE001 - interpreter teardown
            if not self.shutdown_flag:
                raise


# -- async connection hub ----------------------------------------------------

class QueueCommunicator:
    """Hub over a dynamic set of connections.

    Receives are pumped by one daemon thread waiting on the live set;
    sends are drained from a queue by another.  Any IO error retires the
    connection (peers join and leave at any time — the reference's
    elasticity contract, connection.py:198-229).  Per-connection counters
    and an explicit ``close()`` are this implementation's additions.
    """

    _IO_ERRORS = (TimeoutError, ConnectionResetError, BrokenPipeError,
                  EOFError, OSError)

    def __init__(self, conns=()):
        self.input_queue = queue.Queue(maxsize=256)
        self.output_queue = queue.Queue(maxsize=256)
        self._conns = {}                  # conn -> {'sent': n, 'recvd': n}
        self._lock = threading.Lock()
        self._wake = threading.Event()    # set while the live set is nonempty
        self._closed = False
        for conn in conns:
            self.add_connection(conn)
        self._threads = [
            threading.Thread(target=self._pump_in, daemon=True),
            threading.Thread(target=self._pump_out, daemon=True),
        ]
        for t in self._threads:
            t.start()

    # - membership -
    def connection_count(self):
        with self._lock:
            return len(self._conns)

    def add_connection(self, conn):
        with self._lock:
            self._conns[conn] = {'sent': 0, 'recvd': 0}
            self._wake.set()

    def disconnect(self, conn):
        with self._lock:
            if self._conns.pop(conn, None) is not None:
                print('disconnected')
            if not self._conns:
                self._wake.clear()

    def close(self):
        self._closed = True
        with self._lock:
            conns = list(self._conns)
            self._conns.clear()
            self._wake.clear()
        for c in conns:
            try:
                c.close()
            except self._IO_ERRORS:
                pass

    # - messaging -
    def recv(self, timeout=None):
        return self.input_queue.get(timeout=timeout)

    def send(self, conn, send_data):
        self.output_queue.put((conn, send_data))

    # - pumps -
    def _pump_out(self):
        while not self._closed:
            try:
                conn, data = self.output_queue.get(timeout=0.5)
            except queue.Empty:
                continue
            try:
                conn.send(data)
            except self._IO_ERRORS:
                self.disconnect(conn)
                continue
            with self._lock:
                if conn in self._conns:
                    self._conns[conn]['sent'] += 1

    def _pump_in(self):
        while not self._closed:
            if not self._wake.wait(timeout=0.5):
                continue
            with self._lock:
                live = list(self._conns)
            if not live:
                continue
            try:
                ready = mp_connection.wait(live, timeout=0.3)
            except self._IO_ERRORS:
                continue
            for conn in ready:
                try:
                    data = conn.recv()
                except self._IO_ERRORS:
                    self.disconnect(conn)
                    continue
                with self._lock:
                    if conn in self._conns:
                        self._conns[conn]['recvd'] += 1
                self.input_queue.put((conn, data))
