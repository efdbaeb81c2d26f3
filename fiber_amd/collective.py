"""Pool-owned RCCL collective plane.

The north-star data path (SURVEY §2b "MI355X equivalents"): ``Pool.map``
fan-out/fan-in rides RCCL broadcast / reduce over xGMI for tensor
payloads, while scalar metadata keeps riding the shm rings.  This module
provides the communicator machinery:

* :class:`GroupMaster` — master-side: owns the rendezvous (a TCPStore
  hosted in the pool master, which is alive for the pool's whole life)
  and a **generation** counter.  A worker death invalidates the
  communicator; the master bumps the generation, rotates the port, and
  broadcasts a rebuild — the next collective re-initializes cleanly.
* :class:`WorkerGroup` — worker-side: lazily joins the process group for
  the current generation (backend ``"nccl"`` == RCCL over xGMI on GPU
  workers, ``"gloo"`` for CPU plumbing tests) and exposes the same
  collective surface as :class:`fiber_amd.ring.RingContext` so ES engines
  run unchanged inside pool workers.

The reference has no analog — fiber's nanomsg PUSH/PULL moved pickled
bytes only (``/root/reference/fiber/pool.py:906-920``); tensor-aware
fan-out/fan-in is the MI355X-native upgrade BASELINE.json names.
"""

import datetime
import os
import socket as _socket
import threading


def _free_tcp_port():
    s = _socket.socket(_socket.AF_INET, _socket.SOCK_STREAM)
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _pg_timeout():
    # Pool-group default is deliberately shorter than the Ring/bench
    # default (600 s): recovery from a member death converges in a few
    # timeout windows (a rank blocked on a dead generation must time out
    # before it joins the new one), so this bounds the
    # kill-to-recovered latency at ~2-3 minutes worst case.  RCCL init
    # and per-block collectives finish in seconds when healthy.
    return datetime.timedelta(
        seconds=float(os.environ.get("FAM_PG_TIMEOUT", "60"))
    )


_side_stream = None


def stage_to_device(t, device):
    """Host->device staging for shared-tensor fan-out: pinned buffer +
    `hipMemcpyAsync` on a dedicated side stream (the north-star staging
    path) — the copy engine runs the transfer while the
    compute stream keeps working; the caller's stream only waits at the
    end.  Device-resident and CPU-target tensors pass through."""
    global _side_stream
    import torch

    if t.device == device:
        return t.contiguous()
    if device.type != "cuda" or t.device.type != "cpu":
        return t.to(device).contiguous()
    if _side_stream is None:
        _side_stream = torch.cuda.Stream(device=device)
    pinned = t.contiguous().pin_memory()
    out = torch.empty_like(pinned, device=device)
    with torch.cuda.stream(_side_stream):
        out.copy_(pinned, non_blocking=True)
    torch.cuda.current_stream().wait_stream(_side_stream)
    return out


class CollectiveError(RuntimeError):
    """A communicator-level failure (rendezvous / transport), as opposed
    to a user-code exception.  The pool master reacts by rotating the
    group generation so retries start from a virgin namespace instead of
    re-using half-initialized gloo/RCCL state."""


class GroupMaster:
    """Master-side communicator bookkeeping (no rank of its own).

    The master never joins the process group — coordination rides the shm
    rings — so a CPU-only master can drive an all-GPU worker group."""

    def __init__(self, world, backend=None):
        self.world = world
        self.backend = backend  # None: workers decide (nccl on GPU)
        self.gen = 0
        self.host = "127.0.0.1"
        self.port = None
        self._store = None
        # ensure() races between the worker-spawn thread (descriptor())
        # and the submitting thread; a double-create would orphan the
        # store half the workers were told about.
        self._lock = threading.Lock()

    def ensure(self):
        with self._lock:
            if self._store is None:
                self._new_store()
        return self

    def _new_store(self):
        import torch.distributed as dist

        self.port = _free_tcp_port()
        # wait_for_workers=False: the server thread must not block the
        # master; workers rendezvous against it with their own timeout.
        self._store = dist.TCPStore(
            self.host, self.port, self.world, True,
            wait_for_workers=False,
        )

    def rebuild(self):
        """Invalidate the current generation (a member died)."""
        with self._lock:
            self.gen += 1
            self._store = None  # drop the old server; port is rotated
            self._new_store()
        return self.descriptor()

    def descriptor(self, rank=None):
        self.ensure()
        with self._lock:  # snapshot: a concurrent rebuild() must not
            d = {         # yield a (new gen, old port) mix
                "gen": self.gen,
                "world": self.world,
                "host": self.host,
                "port": self.port,
                "backend": self.backend,
            }
        if rank is not None:
            d["rank"] = rank
        return d

    def close(self):
        self._store = None


class WorkerGroup:
    """Per-worker collective context (RingContext-compatible surface)."""

    def __init__(self, desc):
        self._desc = dict(desc)
        self.rank = desc["rank"]
        self._inited_gen = None
        self._store = None
        self._device = None

    # -- membership --------------------------------------------------------
    @property
    def size(self):
        return self._desc["world"]

    world = size

    @property
    def gen(self):
        return self._desc["gen"]

    @property
    def backend(self):
        b = self._desc.get("backend")
        if b is None:
            import torch

            b = "nccl" if torch.cuda.is_available() else "gloo"
        return b

    @property
    def device(self):
        import torch

        if self._device is None:
            if self.backend == "nccl":
                # HIP_VISIBLE_DEVICES pins this worker to one MI355X.
                self._device = torch.device("cuda", 0)
                torch.cuda.set_device(self._device)
            else:
                self._device = torch.device("cpu")
        return self._device

    def apply_rebuild(self, desc):
        """New generation from the master: tear down, re-init lazily."""
        if desc["gen"] <= self._desc["gen"]:
            return
        self.destroy()
        rank = self.rank
        self._desc = dict(desc)
        self._desc["rank"] = rank
        self.rank = rank

    def ensure(self):
        if self._inited_gen == self._desc["gen"]:
            return self
        import torch.distributed as dist

        self.destroy()
        d = self._desc
        self._store = dist.TCPStore(
            d["host"], d["port"], d["world"], False, _pg_timeout()
        )
        prefixed = dist.PrefixStore("famgen%d" % d["gen"], self._store)
        # init_process_group namespaces its store keys with a
        # PROCESS-LOCAL group counter; after failed attempts the counter
        # skews across ranks and they rendezvous on different keys
        # forever.  This process owns only the pool group, so pin the
        # counter: every rank of a generation meets at /famgenG/0/.
        try:
            from torch.distributed.distributed_c10d import _world

            _world.group_count = 0
        except Exception:  # pragma: no cover - private API moved
            pass
        try:
            dist.init_process_group(
                backend=self.backend,
                store=prefixed,
                rank=self.rank,
                world_size=d["world"],
                timeout=_pg_timeout(),
            )
        except Exception as exc:
            self.destroy()
            raise CollectiveError(
                "group init failed (gen %d): %s" % (d["gen"], exc)
            ) from exc
        self._inited_gen = d["gen"]
        _ = self.device
        return self

    def _collective(self, op):
        """Run one collective; a transport failure resets local state and
        surfaces as CollectiveError (master rotates the generation)."""
        self.ensure()
        try:
            return op()
        except Exception as exc:
            self.destroy()
            raise CollectiveError(
                "collective failed (gen %d): %s" % (self._desc["gen"], exc)
            ) from exc

    def destroy(self):
        import torch.distributed as dist

        if self._inited_gen is not None and dist.is_initialized():
            try:
                dist.destroy_process_group()
            except Exception:
                pass
        self._inited_gen = None
        self._store = None

    # -- collectives (RingContext-compatible) ------------------------------
    def allreduce(self, tensor, average=False):
        import torch.distributed as dist

        def op():
            dist.all_reduce(tensor, op=dist.ReduceOp.SUM)
            if average:
                tensor.div_(self.size)
            return tensor

        return self._collective(op)

    def broadcast(self, tensor, src=0):
        import torch.distributed as dist

        return self._collective(
            lambda: (dist.broadcast(tensor, src=src), tensor)[1]
        )

    def all_gather(self, tensor):
        import torch
        import torch.distributed as dist

        def op():
            out = [torch.empty_like(tensor) for _ in range(self.size)]
            dist.all_gather(out, tensor)
            return out

        return self._collective(op)

    def all_gather_into(self, out, tensor):
        import torch.distributed as dist

        return self._collective(
            lambda: (dist.all_gather_into_tensor(out, tensor), out)[1]
        )

    def reduce_scatter(self, out, tensor):
        import torch.distributed as dist

        return self._collective(
            lambda: (dist.reduce_scatter_tensor(out, tensor), out)[1]
        )

    def barrier(self):
        import torch.distributed as dist

        return self._collective(lambda: dist.barrier())
