"""Mode-string sockets over the C++ shm ring engine.

Mirrors the reference's swappable-transport seam (uber/fiber
``fiber/socket.py:323-425``: modes ``"r"/"w"/"rw"/"req"/"rep"``) with
single-node shared-memory semantics:

* an **address** is a short string (the shm segment base name) — it is
  picklable and re-dialable from any process on the node (the reference's
  ``tcp://host:port`` trick, ``fiber/queues.py:122-137``);
* ``"w"``/``"r"`` — many-producer / many-consumer ends of one MPMC ring
  (push/pull; demand-driven consumption gives at-least-nanomsg fairness);
* ``"rw"`` — bidirectional pair of rings (Pipe);
* ``"req"``/``"rep"`` — credit/request channel: REQ pushes
  ``ident``-enveloped requests on a shared ring and receives on a private
  per-ident reply ring; REP pops requests and replies by ident (what the
  resilient pool's pending-table attribution rides on).
"""

import atexit
import os
import struct
import threading

from . import config as fam_config
from . import util

try:
    from . import _transport
except ImportError as exc:  # pragma: no cover
    raise ImportError(
        "fiber_amd._transport extension is not built; run "
        "`python setup.py build_ext --inplace`"
    ) from exc

ShmRing = _transport.ShmRing

_owned_rings = []
_owned_lock = threading.Lock()


def _register_owned(ring):
    with _owned_lock:
        _owned_rings.append(ring)


@atexit.register
def _cleanup_owned():
    with _owned_lock:
        for ring in _owned_rings:
            try:
                ring.close()
                ring.unlink()
            except Exception:
                pass
        _owned_rings.clear()


def new_address(prefix="fam-ch"):
    return util.random_name(prefix)


class _RecvBuffers(threading.local):
    """Per-thread reusable receive buffer for single-copy ring reads."""

    def __init__(self):
        self.buf = bytearray(64 << 10)


_recv_buffers = _RecvBuffers()


def ring_recv_view(ring, timeout):
    """Single-copy receive: returns a memoryview into a thread-local
    buffer (valid until the next recv on this thread), or None on
    timeout."""
    buf = _recv_buffers.buf
    while True:
        n = ring.recv_into(buf, timeout)
        if n >= 0:
            return memoryview(buf)[:n]
        if n == -1:
            return None
        need = -(n + 2)  # buffer too small: message is `need` bytes
        _recv_buffers.buf = buf = bytearray(max(need, 2 * len(buf)))


def _capacity():
    return fam_config.get_object().ring_capacity


def _open(name, create, capacity=None, open_timeout=20.0):
    ring = ShmRing(name, create, capacity or _capacity(), open_timeout)
    if create:
        _register_owned(ring)
    return ring


class Socket:
    """A mode-string socket bound or connected to an address."""

    def __init__(self, mode, addr=None, bind=False, capacity=None, ident=None):
        if mode not in ("r", "w", "rw", "req", "rep"):
            raise ValueError("bad socket mode %r" % mode)
        self.mode = mode
        self.addr = addr or new_address()
        self.bound = bind
        cap = capacity
        self._rings = {}

        if mode in ("r", "w"):
            # One shared MPMC ring.  The binder creates it.
            self._rings["main"] = _open(self.addr, bind, cap)
        elif mode == "rw":
            # .a flows binder->dialer, .b flows dialer->binder.
            self._rings["a"] = _open(self.addr + ".a", bind, cap)
            self._rings["b"] = _open(self.addr + ".b", bind, cap)
        elif mode == "rep":
            if not bind:
                raise ValueError("'rep' sockets must bind")
            self._rings["q"] = _open(self.addr + ".q", True, cap)
            self._reply_cache = {}
        elif mode == "req":
            self.ident = ident or util.random_name("i")[:24]
            self._rings["q"] = _open(self.addr + ".q", False, cap)
            self._rings["r"] = _open(
                self.addr + ".r." + self.ident, True, cap
            )

    # -- plain r/w ---------------------------------------------------------
    def send(self, data, timeout=-1.0):
        if self.mode == "w":
            return self._rings["main"].send(data, timeout)
        if self.mode == "rw":
            ring = self._rings["b"] if not self.bound else self._rings["a"]
            return ring.send(data, timeout)
        if self.mode == "req":
            ident = self.ident.encode()
            envelope = struct.pack(">B", len(ident)) + ident + bytes(data)
            return self._rings["q"].send(envelope, timeout)
        raise ValueError("socket mode %r cannot send()" % self.mode)

    def _recv_ring(self):
        if self.mode == "r":
            return self._rings["main"]
        if self.mode == "rw":
            return self._rings["a"] if not self.bound else self._rings["b"]
        if self.mode == "req":
            return self._rings["r"]
        raise ValueError("socket mode %r cannot recv()" % self.mode)

    def recv(self, timeout=-1.0):
        return self._recv_ring().recv(timeout)

    def recv_view(self, timeout=-1.0):
        """Single-copy receive into a thread-local buffer (memoryview
        valid until this thread's next recv_view)."""
        return ring_recv_view(self._recv_ring(), timeout)

    # -- rep ---------------------------------------------------------------
    def recv_request(self, timeout=-1.0):
        """REP side: returns (ident, payload) or None on timeout."""
        data = self._rings["q"].recv(timeout)
        if data is None:
            return None
        (ilen,) = struct.unpack(">B", data[:1])
        ident = data[1 : 1 + ilen].decode()
        return ident, data[1 + ilen :]

    def send_reply(self, ident, data, timeout=-1.0):
        ring = self._reply_cache.get(ident)
        if ring is None:
            # Short open timeout: the requester created its reply ring
            # before its first request, so a missing ring means it died.
            ring = _open(self.addr + ".r." + ident, False, open_timeout=2.0)
            self._reply_cache[ident] = ring
        return ring.send(data, timeout)

    def drop_peer(self, ident):
        """REP side: forget a dead requester's reply ring."""
        self._reply_cache.pop(ident, None)

    # -- stats / lifecycle -------------------------------------------------
    @property
    def pending(self):
        if self.mode in ("r", "w"):
            return self._rings["main"].size
        return sum(r.size for r in self._rings.values())

    def close(self):
        # NOTE: no explicit detach() here — another thread may be blocked
        # inside ring.recv() on the same mapping; close() wakes it (closed
        # flag broadcast) and the munmap happens in the ShmRing destructor
        # once no Python reference holds the ring.
        for ring in self._rings.values():
            try:
                if ring.is_owner:
                    ring.close()
                    ring.unlink()
            except Exception:
                pass
        if self.mode == "rep":
            self._reply_cache.clear()
        self._rings = {}

    def __repr__(self):
        return "Socket(mode=%r, addr=%r, bound=%r)" % (
            self.mode,
            self.addr,
            self.bound,
        )
