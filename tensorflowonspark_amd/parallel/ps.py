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


def _f32_to_bf16(arr_f32):
    """fp32 ndarray -> raw bf16 bytes (round-to-nearest-even)."""
    u = arr_f32.view(np.uint32)
    r = (u + 0x7FFF + ((u >> 16) & 1)) >> 16
    return r.astype(np.uint16).tobytes()


def _bf16_to_f32(raw):
    """raw bf16 bytes -> fp32 ndarray."""
    u = np.frombuffer(raw, dtype=np.uint16).astype(np.uint32) << 16
    return u.view(np.float32)


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
      ("push", bucket_id, lr, momentum, wd[, dtype]) + grad -> params
          dtype "f32" (default) or "bf16": wire format of BOTH the pushed
          gradient and the returned parameters (halves the TCP bytes per
          step; master params stay fp32 server-side)
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
                    bid, lr, mu, wd = header[1:5]
                    dtype = header[5] if len(header) > 5 else "f32"
                    if dtype == "bf16":
                        grad = _bf16_to_f32(payload)
                    else:
                        grad = np.frombuffer(payload, dtype=np.float32)
                    with self._lock:
                        p = self._params[bid]
                        m = self._mom[bid]
                        g = grad + wd * p if wd else grad
                        np.multiply(m, mu, out=m)
                        np.add(m, g, out=m)
                        p -= lr * m
                        data = _f32_to_bf16(p) if dtype == "bf16" \
                            else p.tobytes()
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
        # one conversation per SOCKET at a time; different shards proceed in
        # parallel (AsyncSGD's overlapped round trips run on threads)
        self._locks = [threading.Lock() for _ in self.addrs]

    def shard_lock(self, shard):
        return self._locks[shard % len(self.addrs)]

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

    def push_pull(self, bid, grad_np, out_np, lr, momentum=0.9,
                  weight_decay=0.0, wire="f32"):
        """Push a gradient, receive the updated parameters (one round trip).

        ``wire="bf16"`` halves the bytes both ways (grad pushed and params
        returned in bf16; the server's master copy stays fp32)."""
        s = self._sock(self._shard(bid))
        if wire == "bf16":
            _send_msg(s, ("push", bid, lr, momentum, weight_decay, "bf16"),
                      _f32_to_bf16(grad_np))
            _, payload = _recv_msg(s)
            out_np[:] = _bf16_to_f32(payload)
        else:
            _send_msg(s, ("push", bid, lr, momentum, weight_decay),
                      grad_np.tobytes())
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
    from the server).

    ``overlap=True`` (default): each bucket's push/pull round trip starts from
    the grad-ready hook DURING backward — the TCP transfer of early (output-
    side) buckets runs while the GPU is still computing input-side gradients.
    ``step()`` only joins the in-flight round trips and installs the returned
    parameters. Round 1 did a serial post-step loop over all buckets
    (~100 MB/step blocking the GPU). ``wire="bf16"``
    additionally halves the bytes on the wire.
    """

    def __init__(self, engine, client, lr=0.01, momentum=0.9,
                 weight_decay=0.0, wire="bf16", overlap=True):
        import torch
        self.engine = engine
        self.client = client
        self.lr = lr
        self.momentum = momentum
        self.weight_decay = weight_decay
        self.wire = wire
        self.overlap = overlap
        self._cuda = torch.cuda.is_available() and \
            engine._buckets and engine._buckets[0].buffer.is_cuda
        self._threads = {}
        self._host_out = {}
        self._pin = {}
        for i, bucket in enumerate(engine._buckets):
            pf = bucket.param_flat
            assert pf is not None, "AsyncSGD requires flatten_params=True"
            client.init_bucket(i, pf.detach().cpu().float().numpy())
            # start from the server's (first writer's) params
            host = np.empty(pf.numel(), dtype=np.float32)
            client.pull(i, host)
            with_torch_copy(pf, host)
            self._host_out[i] = np.empty(pf.numel(), dtype=np.float32)
            self._pin[i] = torch.empty(bucket.buffer.numel(),
                                       dtype=torch.float32,
                                       pin_memory=self._cuda)
        if overlap:
            self._bucket_idx = {id(b): i
                                for i, b in enumerate(engine._buckets)}
            engine.bucket_ready_cb = self._on_bucket_ready

    def _on_bucket_ready(self, bucket):
        """Grad-ready hook (fires inside backward): stage the bucket's grad
        off-device asynchronously and start its PS round trip on a thread."""
        import torch
        i = self._bucket_idx[id(bucket)]
        ev = None
        if self._cuda:
            stream = self.engine._comm_stream or torch.cuda.current_stream()
            if self.engine._comm_stream is not None:
                stream.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(stream):
                self._pin[i].copy_(bucket.buffer.float(), non_blocking=True)
            ev = torch.cuda.Event()
            ev.record(stream)
        else:
            self._pin[i].copy_(bucket.buffer.float())
        t = threading.Thread(target=self._roundtrip, args=(i, ev), daemon=True)
        t.start()
        self._threads[i] = t

    def _roundtrip(self, i, ev):
        if ev is not None:
            ev.synchronize()
        grad = self._pin[i].numpy()
        # serialize per SHARD socket; different shards overlap
        with self.client.shard_lock(self.client._shard(i)):
            self.client.push_pull(i, grad, self._host_out[i], self.lr,
                                  self.momentum, self.weight_decay,
                                  wire=self.wire)

    @staticmethod
    def _np(t):
        return t.detach().cpu().float().numpy()

    def zero_grad(self):
        self.engine.zero_grad()

    def step(self):
        if self.overlap and self._threads:
            for i, t in sorted(self._threads.items()):
                t.join()
                with_torch_copy(self.engine._buckets[i].param_flat,
                                self._host_out[i])
            self._threads = {}
            return
        for i, bucket in enumerate(self.engine._buckets):
            grad = self._np(bucket.buffer)
            out = self._host_out[i]
            self.client.push_pull(i, grad, out, self.lr, self.momentum,
                                  self.weight_decay, wire=self.wire)
            with_torch_copy(bucket.param_flat, out)


def with_torch_copy(param_flat, host_np):
    import torch
    with torch.no_grad():
        param_flat.copy_(torch.from_numpy(host_np).to(param_flat.device,
                                                      param_flat.dtype))
