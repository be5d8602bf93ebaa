"""Host-side control-plane transport.

Two carriers with one send/recv interface (parity: reference connection.py):
``multiprocessing.Pipe`` locally and length-prefixed pickle over TCP
remotely (4-byte network-order size header).  The bandwidth-critical GPU
paths (gradient all-reduce, model broadcast) do NOT go through here — they
use RCCL over xGMI (handyrl_amd/dist.py); this layer carries job args,
episodes and results (KB-scale messages).
"""

import io
import pickle
import queue
import socket
import struct
import threading
import multiprocessing as mp
import multiprocessing.connection as mp_connection


def send_recv(conn, sdata):
    conn.send(sdata)
    return conn.recv()


class PickledConnection:
    """Pickle messages over a stream socket with 4-byte length framing."""

    def __init__(self, conn):
        self.conn = conn

    def __del__(self):
        self.close()

    def close(self):
        if self.conn is not None:
            self.conn.close()
            self.conn = None

    def fileno(self):
        return self.conn.fileno()

    def _recv_exact(self, size):
        buf = io.BytesIO()
        while size > 0:
            chunk = self.conn.recv(size)
            if len(chunk) == 0:
                raise ConnectionResetError
            size -= len(chunk)
            buf.write(chunk)
        return buf.getvalue()

    def recv(self):
        (size,) = struct.unpack('!i', self._recv_exact(4))
        return pickle.loads(self._recv_exact(size))

    def _send_all(self, buf):
        view = memoryview(buf)
        while view:
            n = self.conn.send(view)
            view = view[n:]

    def send(self, msg):
        payload = pickle.dumps(msg)
        header = struct.pack('!i', len(payload))
        if 0 < len(payload) <= 16384:
            self._send_all(header + payload)   # coalesce small messages
        else:
            self._send_all(header)
            if payload:
                self._send_all(payload)


def open_socket_connection(port, reuse=False):
    sock = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
    sock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR,
                    sock.getsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR) | 1)
    sock.bind(('', int(port)))
    return sock


def accept_socket_connection(sock):
    try:
        conn, _ = sock.accept()
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
    sock = open_socket_connection(port)
    sock.listen(maxsize)
    sock.settimeout(timeout)
    count = 0
    while count < maxsize:
        conn = accept_socket_connection(sock)
        if conn is not None:
            count += 1
        yield conn


def open_multiprocessing_connections(num_process, target, args_func):
    """Spawn ``num_process`` child processes, each holding one end of a duplex
    pipe; return the parent-side connections."""
    parent_conns = []
    for i in range(num_process):
        conn0, conn1 = mp.Pipe(duplex=True)
        mp.Process(target=target, args=args_func(i, conn1)).start()
        conn1.close()
        parent_conns.append(conn0)
    return parent_conns


class MultiProcessJobExecutor:
    """Generic fan-out pool: a sender thread feeds idle workers from a
    generator; a receiver thread collects results into a bounded queue."""

    def __init__(self, func, send_generator, num_workers, postprocess=None):
        self.send_generator = send_generator
        self.postprocess = postprocess
        self.conns = []
        self.waiting_conns = queue.Queue()
        self.output_queue = queue.Queue(maxsize=8)
        self.shutdown_flag = False

        for i in range(num_workers):
            conn0, conn1 = mp.Pipe(duplex=True)
            mp.Process(target=func, args=(conn1, i), daemon=True).start()
            conn1.close()
            self.conns.append(conn0)
            self.waiting_conns.put(conn0)

    def recv(self):
        return self.output_queue.get()

    def start(self):
        threading.Thread(target=self._sender, daemon=True).start()
        threading.Thread(target=self._receiver, daemon=True).start()

    def _sender(self):
        while not self.shutdown_flag:
            data = next(self.send_generator)
            conn = self.waiting_conns.get()
            conn.send(data)

    def _receiver(self):
        while not self.shutdown_flag:
            for conn in mp_connection.wait(self.conns, timeout=0.3):
                data = conn.recv()
                self.waiting_conns.put(conn)
                if self.postprocess is not None:
                    data = self.postprocess(data)
                self.output_queue.put(data)


class QueueCommunicator:
    """Async hub over a set of connections: daemon send/recv threads with
    bounded queues; dead connections are detected by send/recv exceptions and
    dropped, so peers may join and leave at any time."""

    def __init__(self, conns=()):
        self.input_queue = queue.Queue(maxsize=256)
        self.output_queue = queue.Queue(maxsize=256)
        self.conns = set()
        self._lock = threading.Lock()
        for conn in conns:
            self.add_connection(conn)
        threading.Thread(target=self._send_thread, daemon=True).start()
        threading.Thread(target=self._recv_thread, daemon=True).start()

    def connection_count(self):
        with self._lock:
            return len(self.conns)

    def recv(self, timeout=None):
        return self.input_queue.get(timeout=timeout)

    def send(self, conn, send_data):
        self.output_queue.put((conn, send_data))

    def add_connection(self, conn):
        with self._lock:
            self.conns.add(conn)

    def disconnect(self, conn):
        with self._lock:
            if conn in self.conns:
                print('disconnected')
                self.conns.discard(conn)

    def _send_thread(self):
        while True:
            conn, send_data = self.output_queue.get()
            try:
                conn.send(send_data)
            except (TimeoutError, ConnectionResetError, BrokenPipeError, OSError):
                self.disconnect(conn)

    def _recv_thread(self):
        while True:
            with self._lock:
                conns = list(self.conns)
            if not conns:
                threading.Event().wait(0.1)
                continue
            try:
                ready = mp_connection.wait(conns, timeout=0.3)
            except OSError:
                continue
            for conn in ready:
                try:
                    recv_data = conn.recv()
                except (TimeoutError, ConnectionResetError, EOFError, OSError):
                    self.disconnect(conn)
                    continue
                self.input_queue.put((conn, recv_data))
