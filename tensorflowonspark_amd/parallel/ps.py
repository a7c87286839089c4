"""Asynchronous data parallelism: parameter-server mode.

Capability parity with the reference's ``num_ps``/ParameterServerStrategy path
(reference ``TFCluster.py:225,260-262``, ps lifecycle ``TFSparkNode.py:431-458``,
async-SGD example ``mnist_spark_streaming.py:86``): ``ps``-role executors hold
fp32 master parameter shards; workers push gradients and pull fresh parameters
*without* any cross-worker barrier — each worker advances at its own pace
(bounded only by its own round-trips), which is what makes irregular feeds
(e.g. streaming) deadlock-free where sync all-reduce would stall.

MI355X-native design: the server is a host-side TCP service speaking the same
length-prefixed framing as the rendezvous layer but with raw tensor payloads
(no per-element pickling); parameters live as flat fp32 buckets matching the
worker's ``DDPEngine`` bucket layout, so a push/pull is a handful of large
contiguous sends. The optimizer (SGD w/ momentum) runs server-side, as in the
classic PS architecture.
"""

import logging
import socket
import struct
import threading

import numpy as np

logger = logging.getLogger(__name__)


def _send_msg(sock, header, payload=b""):
    """header: small picklable tuple (json-free for bytes); payload: raw."""
    import pickle
    h = pickle.dumps(header)
    sock.sendall(struct.pack(">II", len(h), len(payload)))
    sock.sendall(h)
    if payload:
        sock.sendall(payload)


def _recv_exact(sock, n):
    buf = bytearray()
    while len(buf) < n:
        chunk = sock.recv(n - len(buf))
        if not chunk:
            return None
        buf += chunk
    return bytes(buf)


def _recv_msg(sock):
    import pickle
    hdr = _recv_exact(sock, 8)
    if hdr is None:
        return None, None
    hlen, plen = struct.unpack(">II", hdr)
    h = pickle.loads(_recv_exact(sock, hlen))
    p = _recv_exact(sock, plen) if plen else b""
    return h, p


class ParameterServer:
    """Holds fp32 master copies of a set of parameter buckets.

    Protocol (header tuple, payload raw bytes):
      ("init", bucket_id, nelem)  + fp32 payload -> ack (first writer wins)
      ("pull", bucket_id)                        -> fp32 payload
      ("push", bucket_id, lr, momentum, wd)      + fp32 grad -> fp32 params
      ("stop",)                                  -> ack, server exits
    """

    def __init__(self, port=0, sock=None):
        self._params = {}
        self._mom = {}
        self._lock = threading.Lock()
        self._stop = threading.Event()
        if sock is not None:
            self._listener = sock
        else:
            self._listener = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
            self._listener.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
            self._listener.bind(("", port))
        self._listener.listen(64)
        self.port = self._listener.getsockname()[1]

    def serve_forever(self):
        """Blocking accept loop; returns when a stop message arrives."""
        self._listener.settimeout(1.0)
        threads = []
        while not self._stop.is_set():
            try:
                conn, _ = self._listener.accept()
            except socket.timeout:
                continue
            except OSError:
                break
            t = threading.Thread(target=self._serve_conn, args=(conn,), daemon=True)
            t.start()
            threads.append(t)
        self._listener.close()
        logger.info("parameter server stopped")

    def stop(self):
        self._stop.set()

    def _serve_conn(self, conn):
        try:
            while not self._stop.is_set():
                header, payload = _recv_msg(conn)
                if header is None:
                    return
                op = header[0]
                if op == "init":
                    _, bid, nelem = header
                    with self._lock:
                        if bid not in self._params:
                            self._params[bid] = np.frombuffer(
                                payload, dtype=np.float32).copy()
                            self._mom[bid] = np.zeros(nelem, dtype=np.float32)
                    _send_msg(conn, ("ok",))
                elif op == "pull":
                    _, bid = header
                    with self._lock:
                        data = self._params[bid].tobytes()
                    _send_msg(conn, ("ok",), data)
                elif op == "push":
                    _, bid, lr, mu, wd = header
                    grad = np.frombuffer(payload, dtype=np.float32)
                    with self._lock:
                        p = self._params[bid]
                        m = self._mom[bid]
                        g = grad + wd * p if wd else grad
                        np.multiply(m, mu, out=m)
                        np.add(m, g, out=m)
                        p -= lr * m
                        data = p.tobytes()
                    _send_msg(conn, ("ok",), data)
                elif op == "stop":
                    _send_msg(conn, ("ok",))
                    self._stop.set()
                    return
        except (OSError, ConnectionError):
            return
        finally:
            try:
                conn.close()
            except OSError:
                pass


class PSClient:
    """Worker-side client: shards DDPEngine buckets across ps addresses."""

    def __init__(self, ps_addrs):
        self.addrs = [(h, int(p)) for h, p in
                      (a.rsplit(":", 1) for a in ps_addrs)]
        self._socks = {}

    def _sock(self, shard):
        if shard not in self._socks:
            import time
            last = None
            for attempt in range(30):  # ps may still be binding its port
                try:
                    self._socks[shard] = socket.create_connection(
                        self.addrs[shard], timeout=120)
                    break
                except OSError as e:
                    last = e
                    time.sleep(1)
            else:
                raise ConnectionError("parameter server {} unreachable: {}"
                                      .format(self.addrs[shard], last))
        return self._socks[shard]

    def _shard(self, bid):
        return bid % len(self.addrs)

    def init_bucket(self, bid, params_np):
        s = self._sock(self._shard(bid))
        _send_msg(s, ("init", bid, params_np.size), params_np.tobytes())
        _recv_msg(s)

    def pull(self, bid, out_np):
        s = self._sock(self._shard(bid))
        _send_msg(s, ("pull", bid))
        _, payload = _recv_msg(s)
        out_np[:] = np.frombuffer(payload, dtype=np.float32)

    def push_pull(self, bid, grad_np, out_np, lr, momentum=0.9, weight_decay=0.0):
        """Push a gradient, receive the updated parameters (one round trip)."""
        s = self._sock(self._shard(bid))
        _send_msg(s, ("push", bid, lr, momentum, weight_decay), grad_np.tobytes())
        _, payload = _recv_msg(s)
        out_np[:] = np.frombuffer(payload, dtype=np.float32)

    def stop_all(self):
        for shard in range(len(self.addrs)):
            try:
                s = self._sock(shard)
                _send_msg(s, ("stop",))
                _recv_msg(s)
            except OSError:
                pass

    def close(self):
        for s in self._socks.values():
            try:
                s.close()
            except OSError:
                pass


class AsyncSGD:
    """Optimizer facade over (DDPEngine buckets, PSClient).

    Use with ``DDPEngine(model, broadcast_params=False, flatten_params=True)``
    and *no* process group (each worker runs independently; consistency comes
    from the server). ``step()`` pushes each bucket's gradient and installs the
    returned parameters — overlap is per-bucket (push bucket i while i+1's
    grad is still being copied off-device).
    """

    def __init__(self, engine, client, lr=0.01, momentum=0.9, weight_decay=0.0):
        self.engine = engine
        self.client = client
        self.lr = lr
        self.momentum = momentum
        self.weight_decay = weight_decay
        for i, bucket in enumerate(engine._buckets):
            pf = bucket.param_flat
            assert pf is not None, "AsyncSGD requires flatten_params=True"
            client.init_bucket(i, pf.detach().cpu().float().numpy())
            # start from the server's (first writer's) params
            host = np.empty(pf.numel(), dtype=np.float32)
            client.pull(i, host)
            with_torch_copy(pf, host)

    @staticmethod
    def _np(t):
        return t.detach().cpu().float().numpy()

    def zero_grad(self):
        self.engine.zero_grad()

    def step(self):
        for i, bucket in enumerate(self.engine._buckets):
            grad = self._np(bucket.buffer)
            out = np.empty_like(grad)
            self.client.push_pull(i, grad, out, self.lr, self.momentum,
                                  self.weight_decay)
            with_torch_copy(bucket.param_flat, out)


def with_torch_copy(param_flat, host_np):
    import torch
    with torch.no_grad():
        param_flat.copy_(torch.from_numpy(host_np).to(param_flat.device,
                                                      param_flat.dtype))
