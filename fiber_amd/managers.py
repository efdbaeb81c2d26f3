"""Shared-state managers: a remote object server + proxies.

Capability parity with reference ``fiber/managers.py`` (SyncManager /
AsyncManager / BaseManager / proxies), built from scratch on a Unix-socket
RPC server hosted in a fiber Process (the manager server *is* a job, as in
the reference, ``fiber/managers.py:154-187``):

* ``SyncManager`` registers dict / list / Namespace / Value / Array /
  Queue / JoinableQueue (Lock/Semaphore/Event are *not* supported — the
  reference comments them out, ``fiber/managers.py:624-633``);
* ``AsyncManager`` provides fire-now-collect-later RPC: ``proxy.method()``
  returns an :class:`AsyncProxyResult` handle; ``.get()`` collects.

Wire protocol per connection (4-byte BE length + pickle):
  ("create", typeid, args, kwds)        -> ("ok", (refid, exposed))
  ("call", refid, method, args, kwds)   -> ("ok", value) | ("error", exc)
  ("snapshot", refid)                   -> ("ok", list(obj))
  ("decref", refid)                     -> ("ok", None)
  ("shutdown",)                         -> ("ok", None)
"""

import os
import queue as _stdlib_queue
import socket
import threading

from . import serialization, util
from .popen import recv_msg, send_msg
from .process import Process
from .queues import Pipe


# ---------------------------------------------------------------------------
# Server-side managed types
# ---------------------------------------------------------------------------


class Namespace:
    def __init__(self, **kwds):
        self.__dict__.update(kwds)

    def __repr__(self):
        items = ", ".join(
            "%s=%r" % kv for kv in sorted(self.__dict__.items())
        )
        return "Namespace(%s)" % items


class _Value:
    def __init__(self, typecode, value):
        self._typecode = typecode
        self._value = value

    def get(self):
        return self._value

    def set(self, value):
        self._value = value


class _Array:
    def __init__(self, typecode, sequence):
        self._typecode = typecode
        self._data = list(sequence)

    def __getitem__(self, i):
        return self._data[i]

    def __setitem__(self, i, v):
        self._data[i] = v

    def __len__(self):
        return len(self._data)

    def tolist(self):
        return list(self._data)


_DEFAULT_REGISTRY = {
    "dict": dict,
    "list": list,
    "Namespace": Namespace,
    "Value": _Value,
    "Array": _Array,
    "Queue": _stdlib_queue.Queue,
    "JoinableQueue": _stdlib_queue.Queue,
}


class Server:
    """Thread-per-client object server on a Unix socket."""

    def __init__(self, address, registry):
        self.address = address
        self.registry = registry
        self.objects = {}
        self._next_ref = 0
        self._lock = threading.Lock()
        self._shutdown = threading.Event()

    def bind(self):
        self._sock = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
        self._sock.bind(self.address)
        self._sock.listen(128)
        self._sock.settimeout(0.2)

    def serve_forever(self):
        if getattr(self, "_sock", None) is None:
            self.bind()
        sock = self._sock
        while not self._shutdown.is_set():
            try:
                conn, _ = sock.accept()
            except socket.timeout:
                continue
            except OSError:
                break
            threading.Thread(
                target=self._serve_client, args=(conn,), daemon=True
            ).start()
        sock.close()
        try:
            os.unlink(self.address)
        except OSError:
            pass

    def _serve_client(self, conn):
        try:
            while True:
                data = recv_msg(conn)
                if data is None:
                    return
                request = serialization.loads(data)
                try:
                    response = ("ok", self._dispatch(request))
                except SystemExit:
                    send_msg(conn, serialization.dumps(("ok", None)))
                    return
                except Exception as exc:  # noqa: BLE001
                    try:
                        serialization.dumps(exc)
                        response = ("error", exc)
                    except Exception:
                        response = ("error", RuntimeError(repr(exc)))
                send_msg(conn, serialization.dumps(response))
        except OSError:
            pass
        finally:
            conn.close()

    def _dispatch(self, request):
        op = request[0]
        if op == "create":
            _, typeid, args, kwds = request
            factory = self.registry[typeid]
            obj = factory(*args, **kwds)
            with self._lock:
                refid = self._next_ref
                self._next_ref += 1
                self.objects[refid] = obj
            exposed = [
                m
                for m in dir(obj)
                if callable(getattr(obj, m))
                and (not m.startswith("_") or m in _EXPOSED_DUNDERS)
            ]
            return refid, exposed
        if op == "call":
            _, refid, method, args, kwds = request
            obj = self.objects[refid]
            return getattr(obj, method)(*args, **kwds)
        if op == "snapshot":
            _, refid = request
            return list(self.objects[refid])
        if op == "decref":
            _, refid = request
            with self._lock:
                self.objects.pop(refid, None)
            return None
        if op == "shutdown":
            self._shutdown.set()
            raise SystemExit
        raise ValueError("unknown manager op %r" % (op,))


_EXPOSED_DUNDERS = {
    "__getitem__",
    "__setitem__",
    "__delitem__",
    "__len__",
    "__contains__",
    "__iadd__",
    "__imul__",
    "__getattr__",
    "__setattr__",
    "__delattr__",
}


def _run_server(address, registry, report_conn):
    server = Server(address, registry)
    server.bind()  # bind BEFORE advertising so clients can always connect
    report_conn.send(address)
    server.serve_forever()


# ---------------------------------------------------------------------------
# Client-side proxies
# ---------------------------------------------------------------------------


class _ManagerConnection:
    """One socket to the server; request/response under a lock."""

    def __init__(self, address):
        self.address = address
        self._sock = None
        self.lock = threading.Lock()

    def ensure(self):
        if self._sock is None:
            self._sock = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
            self._sock.connect(self.address)
        return self._sock

    def request(self, payload):
        with self.lock:
            sock = self.ensure()
            send_msg(sock, serialization.dumps(payload))
            data = recv_msg(sock)
        if data is None:
            raise ConnectionError("manager server closed the connection")
        status, value = serialization.loads(data)
        if status == "error":
            raise value
        return value

    def send_only(self, payload):
        sock = self.ensure()
        send_msg(sock, serialization.dumps(payload))

    def recv_only(self):
        data = recv_msg(self._sock)
        if data is None:
            raise ConnectionError("manager server closed the connection")
        status, value = serialization.loads(data)
        if status == "error":
            raise value
        return value

    def close(self):
        if self._sock is not None:
            try:
                self._sock.close()
            except OSError:
                pass
            self._sock = None


class BaseProxy:
    _async = False

    def __init__(self, address, typeid, refid, exposed):
        object.__setattr__(self, "_address", address)
        object.__setattr__(self, "_typeid", typeid)
        object.__setattr__(self, "_refid", refid)
        object.__setattr__(self, "_exposed", set(exposed))
        object.__setattr__(self, "_conn", _ManagerConnection(address))
        object.__setattr__(self, "_pending", [])

    def _callmethod(self, method, args=(), kwds=None):
        payload = ("call", self._refid, method, tuple(args), kwds or {})
        if not self._async:
            return self._conn.request(payload)
        with self._conn.lock:
            self._conn.send_only(payload)
            result = AsyncProxyResult(self)
            self._pending.append(result)
        return result

    def _snapshot(self):
        return self._conn.request(("snapshot", self._refid))

    def __getattr__(self, name):
        if name.startswith("_"):
            raise AttributeError(name)
        if name not in self._exposed:
            raise AttributeError(
                "%r proxy has no method %r" % (self._typeid, name)
            )

        def call(*args, **kwds):
            return self._callmethod(name, args, kwds)

        call.__name__ = name
        return call

    def __reduce__(self):
        return (
            type(self),
            (self._address, self._typeid, self._refid, sorted(self._exposed)),
        )

    def __repr__(self):
        return "<%s proxy for %s ref=%d at %s>" % (
            type(self).__name__,
            self._typeid,
            self._refid,
            self._address,
        )


class AsyncProxyResult:
    """Handle for a fire-now-collect-later RPC (reference
    fiber/managers.py:433-458)."""

    def __init__(self, proxy):
        self._proxy = proxy
        self._done = False
        self._value = None
        self._error = None

    def _fill_next(self, timeout=None):
        import select
        import time as _time

        conn = self._proxy._conn
        pending = self._proxy._pending
        deadline = None if timeout is None else _time.monotonic() + timeout
        with conn.lock:
            while pending and not self._done:
                if deadline is not None:
                    remaining = deadline - _time.monotonic()
                    if remaining <= 0:
                        raise TimeoutError("async result timed out")
                    # Wait for data outside recv_msg so a hung server
                    # cannot block past the deadline (a timeout mid-frame
                    # would corrupt the stream; select-then-recv keeps the
                    # framing intact).
                    ready, _, _ = select.select(
                        [conn.ensure()], [], [], remaining
                    )
                    if not ready:
                        raise TimeoutError("async result timed out")
                head = pending[0]
                try:
                    value = conn.recv_only()
                    head._value = value
                except Exception as exc:  # noqa: BLE001
                    head._error = exc
                head._done = True
                pending.pop(0)

    def get(self, timeout=None):
        if not self._done:
            self._fill_next(timeout)
        if self._error is not None:
            raise self._error
        return self._value

    def ready(self):
        return self._done


class _SyncDunderProxy(BaseProxy):
    def __getitem__(self, key):
        return self._callmethod("__getitem__", (key,))

    def __setitem__(self, key, value):
        return self._callmethod("__setitem__", (key, value))

    def __delitem__(self, key):
        return self._callmethod("__delitem__", (key,))

    def __len__(self):
        return self._callmethod("__len__")

    def __contains__(self, key):
        return self._callmethod("__contains__", (key,))

    def __iter__(self):
        return iter(self._snapshot())


class NamespaceProxy(BaseProxy):
    def __getattr__(self, name):
        if name.startswith("_"):
            raise AttributeError(name)
        return self._callmethod("__getattribute__", (name,))

    def __setattr__(self, name, value):
        if name.startswith("_"):
            return object.__setattr__(self, name, value)
        return self._callmethod("__setattr__", (name, value))


class ValueProxy(BaseProxy):
    @property
    def value(self):
        return self._callmethod("get")

    @value.setter
    def value(self, v):
        self._callmethod("set", (v,))


_PROXY_TYPES = {
    "dict": _SyncDunderProxy,
    "list": _SyncDunderProxy,
    "Array": _SyncDunderProxy,
    "Namespace": NamespaceProxy,
    "Value": ValueProxy,
}


class _AsyncAutoProxy(BaseProxy):
    _async = True


# ---------------------------------------------------------------------------
# Managers
# ---------------------------------------------------------------------------


class BaseManager:
    _registry = dict(_DEFAULT_REGISTRY)
    _proxy_base = BaseProxy

    def __init__(self):
        self._address = None
        self._process = None
        self._conn = None

    @classmethod
    def register(cls, typeid, callable_obj, proxytype=None):
        # Subclasses get their own registry copy on first register.
        if "_registry" not in cls.__dict__:
            cls._registry = dict(cls._registry)
        cls._registry[typeid] = callable_obj
        if proxytype is not None:
            if "_proxy_types" not in cls.__dict__:
                cls._proxy_types = dict(_PROXY_TYPES)
            cls._proxy_types[typeid] = proxytype

    def start(self):
        from . import config as fam_config

        conf = fam_config.get_object()
        address = util.ipc_path(conf, util.random_name("fam-mgr") + ".sock")
        reader, writer = Pipe(duplex=False)
        self._process = Process(
            target=_run_server,
            args=(address, dict(self._registry), writer),
            name="fam-manager",
        )
        self._process.start()
        self._address = reader.recv(timeout=conf.start_timeout)
        reader.close()
        self._conn = _ManagerConnection(self._address)

    def connect(self, address):
        self._address = address
        self._conn = _ManagerConnection(address)

    @property
    def address(self):
        return self._address

    def _create(self, typeid, *args, **kwds):
        refid, exposed = self._conn.request(("create", typeid, args, kwds))
        proxy_types = getattr(type(self), "_proxy_types", _PROXY_TYPES)
        proxy_cls = proxy_types.get(typeid, self._proxy_base)
        return proxy_cls(self._address, typeid, refid, exposed)

    def __getattr__(self, typeid):
        if typeid.startswith("_"):
            raise AttributeError(typeid)
        if typeid in self._registry:

            def factory(*args, **kwds):
                return self._create(typeid, *args, **kwds)

            factory.__name__ = typeid
            return factory
        raise AttributeError(typeid)

    # explicit factories (so dir() and docs show them)
    def dict(self, *args, **kwds):
        return self._create("dict", *args, **kwds)

    def list(self, *args, **kwds):
        return self._create("list", *args, **kwds)

    def Namespace(self, **kwds):
        return self._create("Namespace", **kwds)

    def Value(self, typecode, value):
        return self._create("Value", typecode, value)

    def Array(self, typecode, sequence):
        return self._create("Array", typecode, sequence)

    def Queue(self, maxsize=0):
        return self._create("Queue", maxsize)

    def JoinableQueue(self, maxsize=0):
        return self._create("JoinableQueue", maxsize)

    def shutdown(self):
        if self._conn is not None:
            try:
                self._conn.request(("shutdown",))
            except (ConnectionError, OSError):
                pass
            self._conn.close()
            self._conn = None
        if self._process is not None:
            self._process.join(10)
            if self._process.is_alive():
                self._process.terminate()
                self._process.join(5)
            self._process = None

    def join(self, timeout=None):
        if self._process is not None:
            self._process.join(timeout)

    def __enter__(self):
        if self._address is None:
            self.start()
        return self

    def __exit__(self, *exc):
        self.shutdown()


class SyncManager(BaseManager):
    pass


class AsyncManager(BaseManager):
    """Every registered method call returns an AsyncProxyResult."""

    _proxy_base = _AsyncAutoProxy

    def _create(self, typeid, *args, **kwds):
        refid, exposed = self._conn.request(("create", typeid, args, kwds))
        return _AsyncAutoProxy(self._address, typeid, refid, exposed)
