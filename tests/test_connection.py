"""Transport framing and hub tests (the reference ships none)."""

import socket
import threading

import pytest

from handyrl_amd.connection import (
    PickledConnection, open_socket_connection, accept_socket_connection,
    connect_socket_connection, QueueCommunicator)


def _free_port():
    s = socket.socket()
    s.bind(('', 0))
    port = s.getsockname()[1]
    s.close()
    return port


def test_framed_pickle_roundtrip():
    port = _free_port()
    server_sock = open_socket_connection(port)
    server_sock.listen(1)

    results = {}

    def server():
        conn = accept_socket_connection(server_sock)
        results['got'] = conn.recv()
        conn.send({'reply': results['got']['x'] * 2})
        # large payload (> coalesce threshold)
        big = conn.recv()
        conn.send(len(big))

    t = threading.Thread(target=server, daemon=True)
    t.start()

    client = connect_socket_connection('127.0.0.1', port)
    client.send({'x': 21, 'data': list(range(100))})
    assert client.recv() == {'reply': 42}
    payload = b'z' * 100000
    client.send(payload)
    assert client.recv() == len(payload)
    t.join(timeout=5)
    assert results['got']['x'] == 21


def test_queue_communicator_pipe():
    import multiprocessing as mp
    conn0, conn1 = mp.Pipe(duplex=True)
    hub = QueueCommunicator([conn0])
    assert hub.connection_count() == 1
    conn1.send(('hello', 1))
    conn, msg = hub.recv(timeout=5)
    assert msg == ('hello', 1)
    hub.send(conn, 'world')
    assert conn1.recv() == 'world'
    hub.disconnect(conn)
    assert hub.connection_count() == 0


def test_queue_communicator_close():
    """close() retires every connection and stops the pumps."""
    import multiprocessing as mp
    conn0, conn1 = mp.Pipe(duplex=True)
    hub = QueueCommunicator([conn0])
    assert hub.connection_count() == 1
    hub.close()
    assert hub.connection_count() == 0
