"""Connections, Pipe and SimpleQueue over the shm transport.

Parity with reference ``fiber/queues.py`` (ZConnection / LazyZConnection /
Pipe / SimpleQueuePush):

* :class:`Connection` is a picklable (mode, addr) pair that (re-)dials on
  unpickle and connects lazily on first use — the property that lets queue
  and pipe ends travel inside task args to remote workers;
* :func:`Pipe` builds duplex / simplex pipes from ring pairs;
* :class:`SimpleQueue` is one MPMC ring: many writers, many readers, with
  demand-driven fair consumption.

Objects are serialized with :mod:`fiber_amd.serialization`, so CUDA
tensors put on a queue cross as HIP IPC handles, never by value.
"""

import queue as _stdlib_queue

from . import serialization
from .transport import Socket, new_address, ring_recv_view as _ring_recv


class Connection:
    """Picklable, lazily-dialed connection end."""

    def __init__(self, mode, addr, bound=False, lazy=True):
        self._mode = mode
        self._addr = addr
        self._bound = bound
        self._sock = None
        self._buffered = None
        if not lazy:
            self._ensure()

    def _ensure(self):
        if self._sock is None:
            self._sock = Socket(self._mode, self._addr, bind=self._bound)
        return self._sock

    # -- object API --------------------------------------------------------
    def send(self, obj, timeout=-1.0):
        payload = serialization.dumps_closure(obj)
        return self._ensure().send(payload, timeout)

    def recv(self, timeout=-1.0):
        if self._buffered is not None:
            data, self._buffered = self._buffered, None
            return serialization.loads(data)
        data = self._ensure().recv_view(timeout)
        if data is None:
            raise TimeoutError("recv timed out")
        return serialization.loads(data)

    def send_bytes(self, data, timeout=-1.0):
        return self._ensure().send(data, timeout)

    def recv_bytes(self, timeout=-1.0):
        if self._buffered is not None:
            data, self._buffered = self._buffered, None
            return data
        data = self._ensure().recv(timeout)
        if data is None:
            raise TimeoutError("recv timed out")
        return data

    def poll(self, timeout=0.0):
        """True if a message is available within *timeout* seconds.

        ``poll(None)`` blocks until a message arrives (stdlib
        ``multiprocessing.Connection.poll`` semantics); ``poll()`` /
        ``poll(0)`` is non-blocking."""
        if self._buffered is not None:
            return True
        ring_timeout = -1.0 if timeout is None else float(timeout)
        data = self._ensure().recv(ring_timeout)
        if data is None:
            return False
        self._buffered = data
        return True

    def close(self):
        if self._sock is not None:
            self._sock.close()
            self._sock = None

    @property
    def addr(self):
        return self._addr

    def __del__(self):
        if self._bound:
            try:
                self.close()
            except Exception:
                pass

    # -- pickling: re-dial on the far side (never re-create) ---------------
    def __reduce__(self):
        return (Connection, (self._mode, self._addr, False, True))

    def __repr__(self):
        return "Connection(mode=%r, addr=%r, bound=%r)" % (
            self._mode,
            self._addr,
            self._bound,
        )


def Pipe(duplex=True):
    """Returns a pair of connected Connection objects.

    duplex=True: both ends send and recv.  duplex=False: returns
    (reader, writer).
    """
    addr = new_address("fam-pipe")
    if duplex:
        end_a = Connection("rw", addr, bound=True, lazy=False)
        end_b = Connection("rw", addr, bound=False, lazy=True)
        return end_a, end_b
    reader = Connection("r", addr, bound=True, lazy=False)
    writer = Connection("w", addr, bound=False, lazy=True)
    return reader, writer


class SimpleQueue:
    """MPMC queue usable from any process on the node."""

    def __init__(self, addr=None, _create=None):
        create = _create if _create is not None else (addr is None)
        self._addr = addr or new_address("fam-q")
        self._bound = create
        self._sock = None
        if create:
            # Create the segment eagerly so readers/writers in children
            # can open it regardless of who touches the queue first.
            self._ensure()

    def _ensure(self):
        if self._sock is None:
            # One ring; both put and get use it regardless of bind side.
            mode = "r" if self._bound else "w"
            self._sock = Socket(mode, self._addr, bind=self._bound)
            # Socket.send/recv check mode; use the raw ring directly so one
            # handle serves both directions.
            self._ring = self._sock._rings["main"]
        return self._ring

    def put(self, obj, timeout=-1.0):
        ring = self._ensure()
        payload = serialization.dumps_closure(obj)
        ok = ring.send(payload, timeout)
        if not ok:
            raise TimeoutError("queue put timed out")

    def get(self, timeout=-1.0):
        ring = self._ensure()
        data = _ring_recv(ring, timeout)
        if data is None:
            raise TimeoutError("queue get timed out")
        return serialization.loads(data)

    def put_many(self, objs, timeout=-1.0):
        """Enqueue a batch of objects with one lock hold + one wake per
        burst (~1.7x the per-message rate of put() for small payloads;
        see profiles).  Atomicity is per-message, not per-batch."""
        ring = self._ensure()
        payloads = [serialization.dumps_closure(o) for o in objs]
        sent = ring.send_many(payloads, timeout)
        if sent != len(payloads):
            raise TimeoutError(
                "queue put_many timed out after %d of %d" %
                (sent, len(payloads))
            )

    def get_many(self, max_n=64, timeout=-1.0):
        """Dequeue up to max_n objects; blocks (per timeout) only for the
        first.  Returns a possibly-empty list on timeout."""
        ring = self._ensure()
        return [serialization.loads(d) for d in ring.recv_many(max_n, timeout)]

    def get_nowait(self):
        ring = self._ensure()
        data = _ring_recv(ring, 0.0)
        if data is None:
            raise _stdlib_queue.Empty
        return serialization.loads(data)

    def empty(self):
        return self._ensure().size == 0

    def qsize(self):
        return self._ensure().size

    def close(self):
        if self._sock is not None:
            self._sock.close()
            self._sock = None

    def __reduce__(self):
        return (SimpleQueue, (self._addr, False))

    def __del__(self):
        if self._bound:
            try:
                self.close()
            except Exception:
                pass

    def __repr__(self):
        return "SimpleQueue(addr=%r)" % self._addr
