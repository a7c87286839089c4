"""Driver<->executor rendezvous over TCP.

Capability parity with reference ``tensorflowonspark/reservation.py`` (REG/QUERY/
QINFO/STOP message semantics, env-var bind overrides ``TFOS_SERVER_HOST``/
``TFOS_SERVER_PORT`` incl. ``lo-hi`` port ranges, 600 s await timeout, 1 Hz client
polling, reference ``reservation.py:100-301``) — new design decisions:

* Frames are 4-byte big-endian length-prefixed **JSON** (not pickle): the roster is
  pure metadata and a JSON control plane is language-agnostic, so the native C++
  inference CLI and any JVM integration can speak it without Python.
* The completed roster is what seeds ``torch.distributed`` RCCL init: rank =
  position of the node's ``executor_id`` in the sorted roster, and the
  chief/master node's (host, port) becomes MASTER_ADDR/MASTER_PORT.
"""

import json
import logging
import os
import select
import socket
import struct
import threading
import time

logger = logging.getLogger(__name__)

TFOS_SERVER_HOST = "TFOS_SERVER_HOST"
TFOS_SERVER_PORT = "TFOS_SERVER_PORT"

#: seconds to wait for every executor to register before declaring the job dead
DEFAULT_TIMEOUT = 600


class Reservations:
    """Thread-safe store of registered node metadata dicts."""

    def __init__(self, required):
        self.required = required
        self._lock = threading.Condition()
        self._nodes = []
        self._stopped = False

    def add(self, meta):
        with self._lock:
            self._nodes.append(meta)
            self._lock.notify_all()

    def done(self):
        with self._lock:
            return self._stopped or len(self._nodes) >= self.required

    def get(self):
        with self._lock:
            return list(self._nodes)

    def remaining(self):
        with self._lock:
            return self.required - len(self._nodes)

    def stop(self):
        with self._lock:
            self._stopped = True
            self._lock.notify_all()


class MessageSocket:
    """Length-prefixed JSON framing over a stream socket."""

    def receive(self, sock):
        header = self._recv_exact(sock, 4)
        if header is None:
            return None
        (length,) = struct.unpack(">I", header)
        payload = self._recv_exact(sock, length)
        if payload is None:
            return None
        return json.loads(payload.decode("utf-8"))

    def send(self, sock, msg):
        payload = json.dumps(msg).encode("utf-8")
        sock.sendall(struct.pack(">I", len(payload)) + payload)

    @staticmethod
    def _recv_exact(sock, n):
        buf = b""
        while len(buf) < n:
            chunk = sock.recv(n - len(buf))
            if not chunk:
                return None
            buf += chunk
        return buf


class Server(MessageSocket):
    """Rendezvous server run on the driver.

    Handles: ``REG`` (register node meta), ``QUERY`` (are all nodes present),
    ``QINFO`` (return roster), ``STOP`` (mark done/stopped) — the same message set
    as reference ``reservation.py:130-146``.
    """

    def __init__(self, count):
        self.reservations = Reservations(count)
        self.done = threading.Event()
        self._listener = None
        self._thread = None
        self._done_nodes = []
        self._done_lock = threading.Lock()

    # -- lifecycle -----------------------------------------------------------

    def start(self):
        """Bind (honoring env overrides) and start the select() loop thread.

        Returns (host, port).
        """
        host = os.environ.get(TFOS_SERVER_HOST)
        port_spec = os.environ.get(TFOS_SERVER_PORT)
        sock = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        sock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        bind_host = host if host else ""
        if port_spec:
            last_err = None
            for port in _parse_port_spec(port_spec):
                try:
                    sock.bind((bind_host, port))
                    last_err = None
                    break
                except OSError as e:
                    last_err = e
            if last_err is not None:
                raise last_err
        else:
            sock.bind((bind_host, 0))
        sock.listen(64)
        self._listener = sock

        addr_host = host if host else _default_ip()
        addr = (addr_host, sock.getsockname()[1])
        logger.info("reservation server listening on %s", addr)

        self._thread = threading.Thread(target=self._serve, daemon=True)
        self._thread.start()
        return addr

    def done_count(self):
        with self._done_lock:
            return len(self._done_nodes)

    def stop(self):
        self.done.set()
        self.reservations.stop()
        if self._listener is not None:
            try:
                self._listener.close()
            except OSError:
                pass

    # -- driver-side wait ----------------------------------------------------

    def await_reservations(self, sc=None, status=None, timeout=DEFAULT_TIMEOUT):
        """Block until all nodes registered; poll 1 Hz.

        ``status`` is a dict observed for an ``'error'`` key set by the background
        cluster-start thread (reference ``reservation.py:113-128``); on error the
        Spark job is cancelled (if ``sc`` given) and the error re-raised.
        """
        deadline = time.time() + timeout
        while not self.reservations.done():
            if status and status.get("error"):
                if sc is not None:
                    try:
                        sc.cancelAllJobs()
                    except Exception:  # pragma: no cover - spark-only path
                        pass
                raise RuntimeError("cluster startup failed: {}".format(status["error"]))
            if time.time() > deadline:
                raise TimeoutError(
                    "timed out waiting for {} more reservations after {}s".format(
                        self.reservations.remaining(), timeout))
            logger.info("waiting for %d reservations", self.reservations.remaining())
            time.sleep(1)
        return self.reservations.get()

    # -- internals -----------------------------------------------------------

    def _serve(self):
        conns = [self._listener]
        while not self.done.is_set():
            try:
                readable, _, _ = select.select(conns, [], [], 1)
            except OSError:
                break
            for s in readable:
                if s is self._listener:
                    try:
                        conn, _ = self._listener.accept()
                        conns.append(conn)
                    except OSError:
                        pass
                else:
                    try:
                        msg = self.receive(s)
                    except (OSError, ValueError):
                        msg = None
                    if msg is None:
                        conns.remove(s)
                        s.close()
                        continue
                    self._handle(s, msg)
        for s in conns:
            try:
                s.close()
            except OSError:
                pass

    def _handle(self, sock, msg):
        mtype = msg.get("type")
        if mtype == "REG":
            self.reservations.add(msg["data"])
            self.send(sock, {"type": "OK"})
        elif mtype == "QUERY":
            self.send(sock, {"type": "RESP", "data": self.reservations.done()})
        elif mtype == "QINFO":
            self.send(sock, {"type": "RESP", "data": self.reservations.get()})
        elif mtype == "DONE":
            # a worker's map_fun returned (InputMode.TENSORFLOW completion
            # signal — the analog of the reference's statusTracker polling,
            # reference TFCluster.py:154-169)
            with self._done_lock:
                self._done_nodes.append(msg.get("data"))
            self.send(sock, {"type": "OK"})
        elif mtype == "STOP":
            self.reservations.stop()
            self.send(sock, {"type": "OK"})
        else:
            self.send(sock, {"type": "ERR", "data": "unknown message type"})


class Client(MessageSocket):
    """Executor-side client; reconnects up to 3 times per request."""

    RETRIES = 3

    def __init__(self, server_addr):
        self.server_addr = (server_addr[0], int(server_addr[1]))
        self._sock = None
        self._connect()

    def _connect(self):
        self._sock = socket.create_connection(self.server_addr, timeout=30)

    def close(self):
        if self._sock is not None:
            try:
                self._sock.close()
            except OSError:
                pass
            self._sock = None

    def _request(self, msg):
        last = None
        for attempt in range(self.RETRIES + 1):
            try:
                if self._sock is None:
                    self._connect()
                self.send(self._sock, msg)
                resp = self.receive(self._sock)
                if resp is None:
                    raise ConnectionError("server closed connection")
                return resp
            except (OSError, ConnectionError) as e:
                last = e
                self.close()
                time.sleep(min(2 ** attempt, 5))
        raise ConnectionError(
            "reservation server {} unreachable: {}".format(self.server_addr, last))

    # -- API -----------------------------------------------------------------

    def register(self, meta):
        return self._request({"type": "REG", "data": meta})

    def get_reservations(self):
        return self._request({"type": "QINFO"})["data"]

    def await_reservations(self, timeout=DEFAULT_TIMEOUT):
        deadline = time.time() + timeout
        while True:
            if self._request({"type": "QUERY"})["data"]:
                return self.get_reservations()
            if time.time() > deadline:
                raise TimeoutError("await_reservations timed out after {}s".format(timeout))
            time.sleep(1)

    def request_stop(self):
        return self._request({"type": "STOP"})

    def notify_done(self, executor_id):
        """Report this worker's map_fun completion (TENSORFLOW mode)."""
        return self._request({"type": "DONE", "data": executor_id})


def _parse_port_spec(spec):
    """``'9999'`` -> [9999]; ``'9000-9010'`` -> range inclusive."""
    spec = spec.strip()
    if "-" in spec:
        lo, hi = spec.split("-", 1)
        return range(int(lo), int(hi) + 1)
    return [int(spec)]


def _default_ip():
    """Best-effort externally-routable IP (UDP connect trick; no traffic sent)."""
    s = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
    try:
        s.connect(("8.8.8.8", 53))
        return s.getsockname()[0]
    except OSError:
        return "127.0.0.1"
    finally:
        s.close()
