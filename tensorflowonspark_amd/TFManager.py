"""Per-executor IPC hub (capability parity: reference ``tensorflowonspark/TFManager.py``).

A ``multiprocessing.managers.BaseManager`` serving named ``JoinableQueue``s and a
key/value store. The kv store carries ``state`` in {'running','terminating',
'stopped'} plus shared-memory-ring metadata. In the MI355X design the bulk data
path rides the shared-memory block ring (``utils.shmring``); the manager queues
carry block *descriptors* plus the ``control``/``error`` channels.

Design note: the reference registered ``get``/``set`` callables whose return
values came back as AutoProxies (forcing ``str(...)`` repr-comparison hacks at
every call site). Here the kv store is exposed as a *dict proxy* whose method
calls return plain values, wrapped in a picklable :class:`ManagerHandle` with a
clean ``get``/``set``/``get_queue`` surface.

``mode='remote'`` binds all interfaces (driver-reachable, for ps/evaluator
nodes); ``mode='local'`` binds loopback.
"""

import multiprocessing
from multiprocessing.managers import BaseManager

_qdict = {}
_kdict = {}


def _get_queue(qname):
    return _qdict.get(qname)


def _get_kv():
    return _kdict


class TFManager(BaseManager):
    pass


TFManager.register("get_queue", callable=_get_queue)
TFManager.register("get_kv", callable=_get_kv)


class ManagerHandle:
    """Picklable facade over a (started or connected) TFManager."""

    def __init__(self, mgr, address, authkey):
        self._mgr = mgr
        self.address = tuple(address)
        self._authkey = bytes(authkey)
        self._kv = None

    def _kvp(self):
        if self._kv is None:
            self._kv = self._mgr.get_kv()
        return self._kv

    def get_queue(self, qname):
        return self._mgr.get_queue(qname)

    def get(self, key):
        return self._kvp().get(key)

    def set(self, key, value):
        self._kvp().update({key: value})

    def shutdown(self):
        try:
            self._mgr.shutdown()
        except Exception:
            pass

    def __reduce__(self):
        return (connect, (self.address, self._authkey))


def start(authkey, queues, mode="local"):
    """Create and start a manager process serving ``queues`` + the kv store."""
    _qdict.clear()
    _kdict.clear()
    for q in queues:
        _qdict[q] = multiprocessing.JoinableQueue()
    address = ("", 0) if mode == "remote" else ("127.0.0.1", 0)
    mgr = TFManager(address=address, authkey=authkey)
    mgr.start()
    return ManagerHandle(mgr, mgr.address, authkey)


def connect(address, authkey):
    """Connect to a manager at (host, port) with ``authkey`` bytes."""
    m = TFManager(address=tuple(address), authkey=bytes(authkey))
    m.connect()
    return ManagerHandle(m, address, authkey)
