"""Rendezvous server/client tests (shape parity: reference tests/test_reservation.py)."""

import os
import socket
import threading

import pytest

from tensorflowonspark_amd import reservation


def test_reservations_counting():
    r = reservation.Reservations(3)
    assert not r.done()
    assert r.remaining() == 3
    r.add({"node": 1})
    r.add({"node": 2})
    assert not r.done()
    assert r.remaining() == 1
    r.add({"node": 3})
    assert r.done()
    assert len(r.get()) == 3
    assert r.remaining() == 0


def test_server_client_roundtrip():
    server = reservation.Server(1)
    addr = server.start()
    client = reservation.Client(addr)
    meta = {"executor_id": 0, "host": "1.2.3.4", "port": 2222,
            "job_name": "worker", "task_index": 0, "authkey": "aa"}
    client.register(meta)
    roster = client.await_reservations(timeout=10)
    assert roster == [meta]
    client.request_stop()
    client.close()
    server.stop()


def test_server_env_port():
    # find a free port then pin the server to it
    s = socket.socket()
    s.bind(("", 0))
    port = s.getsockname()[1]
    s.close()
    os.environ[reservation.TFOS_SERVER_PORT] = str(port)
    try:
        server = reservation.Server(1)
        addr = server.start()
        assert addr[1] == port
        server.stop()
    finally:
        del os.environ[reservation.TFOS_SERVER_PORT]


def test_server_env_port_range():
    s = socket.socket()
    s.bind(("", 0))
    port = s.getsockname()[1]
    # hold `port` so the range forces the server onto port+1 or port+2
    os.environ[reservation.TFOS_SERVER_PORT] = "{}-{}".format(port, port + 2)
    try:
        server = reservation.Server(1)
        addr = server.start()
        assert port <= addr[1] <= port + 2 and addr[1] != port
        server.stop()
    finally:
        s.close()
        del os.environ[reservation.TFOS_SERVER_PORT]


def test_server_port_exhaustion():
    s = socket.socket()
    s.bind(("", 0))
    port = s.getsockname()[1]
    os.environ[reservation.TFOS_SERVER_PORT] = str(port)
    try:
        with pytest.raises(OSError):
            reservation.Server(1).start()
    finally:
        s.close()
        del os.environ[reservation.TFOS_SERVER_PORT]


def test_concurrent_registration():
    n = 4
    server = reservation.Server(n)
    addr = server.start()
    errors = []

    def reg(i):
        try:
            c = reservation.Client(addr)
            c.register({"executor_id": i, "host": "h", "port": i})
            got = c.await_reservations(timeout=15)
            assert len(got) == n
            c.close()
        except Exception as e:  # pragma: no cover
            errors.append(e)

    threads = [threading.Thread(target=reg, args=(i,)) for i in range(n)]
    for t in threads:
        t.start()
    roster = server.await_reservations(timeout=15)
    for t in threads:
        t.join(timeout=20)
    assert not errors
    assert sorted(m["executor_id"] for m in roster) == list(range(n))
    server.stop()


def test_await_timeout():
    server = reservation.Server(2)
    server.start()
    with pytest.raises(TimeoutError):
        server.await_reservations(timeout=2)
    server.stop()


def test_await_aborts_on_status_error():
    server = reservation.Server(2)
    server.start()
    with pytest.raises(RuntimeError, match="boom"):
        server.await_reservations(status={"error": "boom"}, timeout=10)
    server.stop()
