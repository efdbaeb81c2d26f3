"""Worker pools over the shm transport.

Rebuild target #1: the reference's ``ZPool`` /
``ResilientZPool`` data path (uber/fiber ``fiber/pool.py:644-1692``),
re-based on shared-memory rings:

* :class:`ZPool` — master pushes chunked tasks onto one MPMC task ring;
  workers (fiber Processes, optionally GPU-pinned) pop on demand
  (demand-driven consumption replaces nanomsg PUSH round-robin — strictly
  better load balance for uneven task durations) and push chunked results
  onto a result ring.  Full multiprocessing.Pool API.
* :class:`ResilientZPool` — REQ/REP credit scheduling: a worker *requests*
  each chunk, the master records it in a per-worker pending table, and a
  dead worker's pending chunks are resubmitted after respawn.  Function
  exceptions kill the worker (tasks must be idempotent) and the chunk is
  retried, so probabilistic failures still converge to a complete result.

Collective mode (``collective=True``, ZPool only) adds the RCCL data
plane (SURVEY §2b "MI355X equivalents"; reference fan-out/fan-in:
``fiber/pool.py:906-920``): each worker joins a pool-owned process group
("nccl" == RCCL over xGMI on GPU workers), shared tensor map-args are
IPC-shipped ONCE to rank 0 and broadcast device-to-device, tensor
results can be summed with one all-reduce, and :meth:`ZPool.run_on_all`
runs one SPMD task per worker (the ES-through-pool fan-out).  Control
messages ride per-worker shm rings in a globally consistent order, so
every worker executes the same collective sequence.

Task wire format: ``(seq, base, func_blob, args, starmap, kwds, flags)``
(``flags`` bit 0 = the map has a collective begin record).
Result wire format: ``(seq, base, values, failure, ident)`` where
``failure`` is None or ``(index, exception)``; ``base == -2`` carries a
collective-reduce final tensor.
"""

import itertools
import os
import queue as _stdlib_queue
import threading
import time

from . import config as fam_config
from . import serialization, util
from .process import Process
from .transport import Socket

DEFAULT_CHUNKSIZE = 32
_SENTINEL_SEQ = -1
_REDUCE_BASE = -2          # result record carrying a reduced tensor
_FLAG_COLLECTIVE = 1       # task flags bit: wait for the map's begin ctl


class _ExcInfo:
    """Picklable carrier for a worker-side exception."""

    def __init__(self, exc):
        try:
            serialization.dumps(exc)
            self.exc = exc
        except Exception:
            self.exc = RuntimeError(repr(exc))
        import traceback

        self.tb = traceback.format_exc()

    def rebuild(self):
        return self.exc


# ---------------------------------------------------------------------------
# Inventory: seq-keyed result collection (reference fiber/pool.py:644-728)
# ---------------------------------------------------------------------------


class Inventory:
    def __init__(self):
        self._lock = threading.Lock()
        self._cond = threading.Condition(self._lock)
        self._jobs = {}
        self._seq = itertools.count()

    # -- dynamic (streaming imap) jobs: total size unknown until the
    # feeder exhausts the iterator; results are freed as consumed -------
    def add_dynamic(self, ordered):
        seq = next(self._seq)
        with self._lock:
            self._jobs[seq] = {
                "dyn": True,
                "ordered": ordered,
                "n": None,           # set when the iterator is exhausted
                "results": {},       # ordered mode: index -> value
                "queue": [],         # unordered mode: FIFO of values
                "arrived": set(),    # dedup for resubmitted chunks
                "yielded": 0,
                "error": None,
            }
        return seq

    def finish_dynamic(self, seq, n):
        with self._cond:
            job = self._jobs.get(seq)
            if job is not None:
                job["n"] = n
            self._cond.notify_all()

    def fail_dynamic(self, seq, exc):
        with self._cond:
            job = self._jobs.get(seq)
            if job is not None and job["error"] is None:
                job["error"] = _ExcInfo(exc)
            self._cond.notify_all()

    def dynamic_backlog(self, seq, base):
        """Items submitted but not yet yielded (feeder back-pressure)."""
        with self._lock:
            job = self._jobs.get(seq)
            if job is None:
                return 0
            return base - job["yielded"]

    def iget_dynamic(self, seq, ordered):
        index = 0
        while True:
            with self._cond:
                job = self._jobs.get(seq)
                if job is None:
                    return
                while True:
                    if ordered and index in job["results"]:
                        value = job["results"].pop(index)
                        break
                    if not ordered and job["queue"]:
                        value = job["queue"].pop(0)
                        break
                    if job["error"] is not None:
                        del self._jobs[seq]
                        raise job["error"].rebuild()
                    if job["n"] is not None and job["yielded"] >= job["n"]:
                        del self._jobs[seq]
                        return
                    self._cond.wait()
                job["yielded"] += 1
            if isinstance(value, _ExcInfo):
                with self._lock:
                    self._jobs.pop(seq, None)
                raise value.rebuild()
            yield value
            index += 1

    def add(self, n):
        seq = next(self._seq)
        with self._lock:
            self._jobs[seq] = {
                "n": n,
                "remaining": n,
                "results": [None] * n,
                "arrived": [False] * n,
                "error": None,
                "order_cursor": 0,
                "unordered": [],
            }
        return seq

    def put(self, seq, base, values, failure):
        with self._cond:
            job = self._jobs.get(seq)
            if job is None:
                return
            if job.get("dyn"):
                for offset, value in enumerate(values):
                    index = base + offset
                    if index in job["arrived"]:
                        continue
                    job["arrived"].add(index)
                    if job["ordered"]:
                        job["results"][index] = value
                    else:
                        job["queue"].append(value)
                if failure is not None:
                    index = base + failure[0]  # failure index is chunk-local
                    exc_info = failure[1]
                    if index not in job["arrived"]:
                        job["arrived"].add(index)
                        if job["ordered"]:
                            job["results"][index] = exc_info
                        else:
                            job["queue"].append(exc_info)
                self._cond.notify_all()
                return
            for offset, value in enumerate(values):
                index = base + offset
                if job["arrived"][index]:
                    continue  # duplicate delivery after a resubmit race
                job["arrived"][index] = True
                job["results"][index] = value
                job["remaining"] -= 1
                job["unordered"].append((index, value))
            if failure is not None:
                index = base + failure[0]  # failure index is chunk-local
                exc_info = failure[1]
                if not job["arrived"][index]:
                    job["arrived"][index] = True
                    job["remaining"] -= 1
                    job["error"] = exc_info
                    job["unordered"].append((index, exc_info))
                    job["results"][index] = exc_info
            self._cond.notify_all()

    def fail_seq(self, seq, exc):
        """Abort one job (e.g. a collective map whose group died)."""
        with self._cond:
            job = self._jobs.get(seq)
            if job is None:
                return
            if job.get("dyn"):
                if job["error"] is None:
                    job["error"] = _ExcInfo(exc)
            elif job["remaining"] > 0:
                job["error"] = _ExcInfo(exc)
                job["remaining"] = 0
            self._cond.notify_all()

    def fail_all(self, exc):
        """Abort every outstanding job (pool terminated)."""
        with self._cond:
            for job in self._jobs.values():
                if job.get("dyn"):
                    if job["error"] is None:
                        job["error"] = _ExcInfo(exc)
                    continue
                if job["remaining"] > 0:
                    job["error"] = _ExcInfo(exc)
                    job["remaining"] = 0
            self._cond.notify_all()

    def done(self, seq):
        with self._lock:
            job = self._jobs.get(seq)
            return job is None or job["remaining"] == 0

    def peek(self, seq):
        """(done, error, results) without consuming (callback watcher)."""
        with self._lock:
            job = self._jobs.get(seq)
            if job is None:
                return True, None, None
            done = job["remaining"] == 0 or job["error"] is not None
            results = list(job["results"]) if done else None
            return done, job["error"], results

    def get(self, seq, timeout=None):
        deadline = None if timeout is None else time.monotonic() + timeout
        with self._cond:
            job = self._jobs[seq]
            while job["remaining"] > 0 and job["error"] is None:
                remaining_t = None
                if deadline is not None:
                    remaining_t = deadline - time.monotonic()
                    if remaining_t <= 0:
                        raise TimeoutError("pool result timed out")
                self._cond.wait(remaining_t)
            if job["error"] is not None:
                raise job["error"].rebuild()
            del self._jobs[seq]
            return job["results"]

    def iget_ordered(self, seq):
        with self._cond:
            job = self._jobs[seq]
        yielded = 0
        while yielded < job["n"]:
            with self._cond:
                while not job["arrived"][yielded] and job["error"] is None:
                    self._cond.wait()
                if job["error"] is not None and not job["arrived"][yielded]:
                    raise job["error"].rebuild()
                value = job["results"][yielded]
            if isinstance(value, _ExcInfo):
                raise value.rebuild()
            yield value
            yielded += 1
        with self._lock:
            self._jobs.pop(seq, None)

    def iget_unordered(self, seq):
        with self._cond:
            job = self._jobs[seq]
        yielded = 0
        while yielded < job["n"]:
            with self._cond:
                while len(job["unordered"]) <= yielded:
                    if job["error"] is not None and job["remaining"] == 0:
                        break
                    self._cond.wait()
                if len(job["unordered"]) <= yielded:
                    raise job["error"].rebuild()
                _, value = job["unordered"][yielded]
            if isinstance(value, _ExcInfo):
                raise value.rebuild()
            yield value
            yielded += 1
        with self._lock:
            self._jobs.pop(seq, None)


class AsyncResult:
    def __init__(self, pool, seq, n, callback=None, error_callback=None,
                 single=False):
        self._pool = pool
        self._seq = seq
        self._n = n
        self._single = single
        self._callback = callback
        self._error_callback = error_callback

    def get(self, timeout=None):
        # callbacks fire from the pool's result thread on completion
        # (mp.Pool semantics), not here
        results = self._pool._inventory.get(self._seq, timeout)
        first_exc = next(
            (r for r in results if isinstance(r, _ExcInfo)), None
        )
        if first_exc is not None:
            raise first_exc.rebuild()
        return results[0] if self._single else results

    def _fire_callbacks(self, error, results):
        """Called once by the pool's result thread when the job is done."""
        try:
            first_exc = error or next(
                (r for r in results if isinstance(r, _ExcInfo)), None
            )
            if first_exc is not None:
                if self._error_callback:
                    self._error_callback(first_exc.rebuild())
            elif self._callback:
                self._callback(results[0] if self._single else results)
        except Exception:  # noqa: BLE001 - callback errors must not kill
            import traceback

            traceback.print_exc()
        finally:
            self._callback = None
            self._error_callback = None

    def wait(self, timeout=None):
        try:
            self._pool._inventory.get(self._seq, timeout)
        except TimeoutError:
            pass

    def ready(self):
        return self._pool._inventory.done(self._seq)

    def successful(self):
        if not self.ready():
            raise ValueError("result not ready")
        job = self._pool._inventory._jobs.get(self._seq)
        if job is None:
            return True  # already collected without error
        return job["error"] is None and not any(
            isinstance(r, _ExcInfo) for r in job["results"]
        )


MapResult = AsyncResult
ApplyResult = AsyncResult


class ReduceResult:
    """Result of ``map(..., reduce='sum')``: ONE tensor — the sum of
    every task's return value — produced worker-side by the pool-wide
    RCCL all-reduce (fan-in over xGMI), shipped to the master once."""

    def __init__(self, pool, seq):
        self._pool = pool
        self._seq = seq

    def get(self, timeout=None):
        holder = self._pool._reduce_results.get(self._seq)
        if holder is None:
            raise RuntimeError("reduce result already collected")
        if not holder[0].wait(timeout):
            raise TimeoutError("reduce result timed out")
        self._pool._reduce_results.pop(self._seq, None)
        try:  # free the placeholder inventory slot
            self._pool._inventory.get(self._seq, 0.001)
        except Exception:
            pass
        if holder[2] is not None:
            raise holder[2]
        return holder[1]

    def wait(self, timeout=None):
        holder = self._pool._reduce_results.get(self._seq)
        if holder is not None:
            holder[0].wait(timeout)

    def ready(self):
        holder = self._pool._reduce_results.get(self._seq)
        return holder is None or holder[0].is_set()


# ---------------------------------------------------------------------------
# Worker side
# ---------------------------------------------------------------------------


def _execute_chunk(func, args, starmap, kwds):
    values = []
    for index, arg in enumerate(args):
        try:
            if starmap:
                values.append(func(*arg, **(kwds or {})))
            elif kwds:  # shared tensors arrive as keyword args
                values.append(func(arg, **kwds))
            else:
                values.append(func(arg))
        except Exception as exc:  # noqa: BLE001
            return values, (index, _ExcInfo(exc))
    return values, None


_current_worker_group = None
_current_coll_state = None


def current_worker_group():
    """Inside a collective pool worker: the :class:`WorkerGroup` (rank,
    size, allreduce/broadcast/all_gather/... over the pool's RCCL
    communicator).  None outside collective workers."""
    return _current_worker_group


class _WorkerCollState:
    """Worker-side collective bookkeeping: the group, per-map shared
    tensors and reduce partials, and the ctl-ring processor.  Control
    messages are executed strictly in ring order — the master sends them
    to every worker in the same order under one lock, which is what
    keeps collective calls matched across ranks."""

    def __init__(self, ctl_addr, group_desc, result_sock, ident):
        from .collective import WorkerGroup

        self.sock = Socket("r", ctl_addr, bind=False)
        desc = dict(group_desc)
        # Seqs of collective maps that failed before this worker spawned
        # (their orphan chunks may still sit in the task ring).
        self.dropped = set(desc.pop("dropped_seqs", ()))
        self.group = WorkerGroup(desc)
        self.result_sock = result_sock
        self.ident = ident
        self.maps = {}  # seq -> {shared, reduce, spec, partial}
        global _current_worker_group, _current_coll_state
        _current_worker_group = self.group
        _current_coll_state = self  # introspection (tests/debug)

    # -- ctl processing ----------------------------------------------------
    def drain(self, timeout=0.0):
        """Process pending ctl messages.  Returns False when the ctl ring
        is gone (pool teardown) — the worker loop should exit.

        The whole backlog is swept first so that control-plane records
        (rebuild / drop — monotonic and collective-free) apply BEFORE any
        queued collective op: a `begin` for a map that failed while this
        worker was booting must be skipped, not staged against a dead
        communicator generation (that was a live cascade: stage ->
        connect to a dropped store -> worker death -> another rebuild)."""
        msgs = []
        while True:
            try:
                msg = self.sock.recv(timeout if not msgs else 0.0)
            except RuntimeError:
                return False
            if msg is None:
                break
            msgs.append(serialization.loads(msg))
            timeout = 0.0
        pending = []
        for msg in msgs:
            if msg[0] == "rebuild":
                self.group.apply_rebuild(msg[1])
            elif msg[0] == "drop":
                self.maps.pop(msg[1], None)
                self.dropped.add(msg[1])
            else:
                pending.append(msg)
        for msg in pending:  # collective ops keep their FIFO order
            if msg[0] in ("begin", "reduce_go") and msg[1] in self.dropped:
                continue
            try:
                self._handle(msg)
            except Exception as exc:  # noqa: BLE001
                # A failed collective must not kill the worker: mark the
                # map dead locally and tell the master (usually the
                # master already failed it via the rebuild path).
                util.get_logger().warning(
                    "collective ctl op %r failed: %r", msg[0], exc
                )
                seq = msg[1] if len(msg) > 1 else None
                if seq is not None:
                    self.maps.pop(seq, None)
                    self.dropped.add(seq)
                    record = (seq, _REDUCE_BASE, [],
                              (self.group.gen, _ExcInfo(exc)), self.ident)
                    try:
                        self.result_sock.send(
                            serialization.dumps(record), timeout=1.0
                        )
                    except Exception:  # noqa: BLE001
                        pass
        return True

    def wait_map(self, seq):
        """Block until the begin record for ``seq`` has been processed
        (it was ctl-broadcast before any of the map's chunks).  Returns
        None for a dropped map (failed collective) — the chunk is an
        orphan and must be skipped."""
        deadline = time.monotonic() + float(
            os.environ.get("FAM_CTL_WAIT", "60")
        )
        while seq not in self.maps:
            if seq in self.dropped:
                return None
            if not self.drain(timeout=1.0):
                raise RuntimeError("pool ctl channel closed")
            if time.monotonic() > deadline:
                # self-healing fallback: treat as dropped rather than
                # wedging the worker forever
                self.dropped.add(seq)
                return None
        return self.maps[seq]

    def _handle(self, msg):
        op = msg[0]
        if op == "begin":
            _, seq, meta, blob, reduce_mode, spec = msg
            shared = self._stage_shared(meta, blob) if meta else {}
            self.maps[seq] = {
                "shared": shared,
                "reduce": reduce_mode,
                "spec": spec,
                "partial": None,
            }
        elif op == "reduce_go":
            _, seq = msg
            self._finish_reduce(seq)
        elif op == "exec":
            _, seq, func_blob, args, kwds = msg
            self._exec(seq, func_blob, args, kwds)
        # "rebuild"/"drop" are consumed by drain() before this point

    def _stage_shared(self, meta, blob):
        """Stage the map's shared tensors: rank 0 materializes them from
        the master's one-time blob (HIP IPC handles for device tensors),
        every other rank allocates empty buffers, then each tensor is
        broadcast over the pool communicator (xGMI on GPU workers) —
        ONE host->device hand-off total instead of one per chunk."""
        import torch

        from .collective import stage_to_device

        device = self.group.device
        shared = {}
        if self.group.rank == 0:
            src = serialization.loads(blob)
            for name in sorted(meta):
                # pinned + hipMemcpyAsync on a side stream for host
                # payloads; device tensors (IPC handles) pass through
                shared[name] = stage_to_device(src[name], device)
        else:
            for name in sorted(meta):
                shape, dtype_str = meta[name]
                dtype = getattr(torch, dtype_str.split(".")[-1])
                shared[name] = torch.empty(shape, dtype=dtype,
                                           device=device)
        for name in sorted(meta):
            self.group.broadcast(shared[name], src=0)
        return shared

    def accumulate(self, seq, values):
        entry = self.maps.get(seq)
        if entry is None:
            return
        for value in values:
            if entry["partial"] is None:
                entry["partial"] = value.clone()
            else:
                entry["partial"] += value

    def _finish_reduce(self, seq):
        import torch

        entry = self.maps.pop(seq, None)
        if entry is None:
            return
        partial = entry["partial"]
        if partial is None:
            shape, dtype_str = entry["spec"]
            dtype = getattr(torch, dtype_str.split(".")[-1])
            partial = torch.zeros(shape, dtype=dtype,
                                  device=self.group.device)
        self.group.allreduce(partial)
        if self.group.rank == 0:
            record = (seq, _REDUCE_BASE, [partial], None, self.ident)
            self.result_sock.send(serialization.dumps(record), timeout=-1.0)

    def _exec(self, seq, func_blob, args, kwds):
        """SPMD exec: every worker runs the same call once; results land
        at index == rank.  The function may use current_worker_group()
        for its own collectives (the ES-through-pool hot path)."""
        from .collective import CollectiveError

        try:
            func = serialization.loads(func_blob)
            value = func(*args, **(kwds or {}))
            record = (seq, self.group.rank, [value], None, self.ident)
        except CollectiveError as exc:
            # communicator fault, not user code: report with the
            # generation so the master rotates the group
            record = (seq, _REDUCE_BASE, [],
                      (self.group.gen, _ExcInfo(exc)), self.ident)
        except Exception as exc:  # noqa: BLE001
            record = (seq, self.group.rank, [], (0, _ExcInfo(exc)),
                      self.ident)
        self.result_sock.send(serialization.dumps(record), timeout=-1.0)

    def close(self):
        global _current_worker_group
        _current_worker_group = None
        self.group.destroy()
        self.sock.close()


def _pool_worker_loop(
    task_addr, result_addr, resilient, maxtasks, init_blob, ident_prefix,
    nproc=1, ctl_addr=None, group_desc=None,
):
    """Worker job entry: runs ``nproc`` worker cores in this job
    (reference multi-worker-per-job, ``cpu_per_job``: forked cores share
    the shm rings; each core gets a derived ident for attribution).
    Forking happens before any socket/HIP state exists in the core."""
    if ctl_addr is not None and nproc > 1:
        raise ValueError("collective mode is incompatible with "
                         "cpu_per_worker > 1 (one rank per process)")
    if nproc > 1:
        children = []
        for k in range(1, nproc):
            pid = os.fork()
            if pid == 0:
                _set_pdeathsig()  # die with the job's primary process
                try:
                    _pool_worker_core(task_addr, result_addr, resilient,
                                      maxtasks, init_blob,
                                      "%s:%d" % (ident_prefix, k))
                except BaseException:
                    # A core fault must take the whole job down so the
                    # master's pending-table resubmission covers every
                    # core's in-flight chunks (the job IS the failure
                    # domain, reference cpu_per_job semantics).
                    import signal as _signal

                    os.kill(os.getppid(), _signal.SIGTERM)
                    os._exit(1)
                os._exit(0)
            children.append(pid)
        try:
            _pool_worker_core(task_addr, result_addr, resilient, maxtasks,
                              init_blob, ident_prefix)
        except BaseException:
            import signal as _signal

            for pid in children:
                try:
                    os.kill(pid, _signal.SIGKILL)
                except OSError:
                    pass
            raise
        finally:
            for pid in children:
                try:
                    os.waitpid(pid, 0)
                except ChildProcessError:
                    pass
        return
    _pool_worker_core(task_addr, result_addr, resilient, maxtasks,
                      init_blob, ident_prefix, ctl_addr, group_desc)


def _set_pdeathsig():
    """Linux: deliver SIGKILL to this process when its parent dies."""
    try:
        import ctypes
        import signal as _signal

        libc = ctypes.CDLL("libc.so.6", use_errno=True)
        libc.prctl(1, _signal.SIGKILL)  # PR_SET_PDEATHSIG
    except Exception:
        pass


def _pool_worker_core(
    task_addr, result_addr, resilient, maxtasks, init_blob, ident_prefix,
    ctl_addr=None, group_desc=None,
):
    """Worker main loop (reference zpool_worker_core, pool.py:760-825)."""
    result_sock = Socket("w", result_addr, bind=False)
    coll = None
    if ctl_addr is not None:
        ident0 = ident_prefix or util.random_name("w")[:24]
        coll = _WorkerCollState(ctl_addr, group_desc, result_sock, ident0)

    if init_blob is not None:
        initializer, initargs = serialization.loads(init_blob)
        initializer(*initargs)

    func_cache = {}
    tasks_done = 0

    ident = ident_prefix or util.random_name("w")[:24]
    if resilient:
        task_sock = Socket("req", task_addr, bind=False, ident=ident)
    else:
        task_sock = Socket("r", task_addr, bind=False)

    while True:
        if coll is not None and not coll.drain():
            break  # ctl ring gone: pool tearing down
        if resilient:
            task_sock.send(b"", timeout=-1.0)
            # Timeout + re-request guards against a discarded request (the
            # master drops requests it cannot attribute to a live worker).
            payload = task_sock.recv_view(timeout=10.0)
        else:
            # Collective workers poll so ctl messages (staging
            # broadcasts, SPMD execs) are served between chunks; 5 ms
            # bounds the ctl latency at ~200 idle wakeups/s.  (SPMD
            # callers amortize further by batching iterations per exec.)
            payload = task_sock.recv_view(
                timeout=0.005 if coll is not None else -1.0
            )
        if payload is None:
            continue
        task = serialization.loads(payload)
        seq, base, func_blob, args, star, kwds, flags = task
        if seq == _SENTINEL_SEQ:
            break
        entry = None
        if flags & _FLAG_COLLECTIVE and coll is not None:
            entry = coll.wait_map(seq)
            if entry is None:
                # orphan chunk of a failed collective map: acknowledge
                # (flow-control) but do not execute
                record = (seq, base, [None] * len(args), None, ident)
                result_sock.send(serialization.dumps(record), timeout=-1.0)
                continue
            if entry["shared"]:
                kwds = dict(kwds or {})
                kwds.update(entry["shared"])
        # Keyed by the blob bytes themselves: a hash collision between two
        # distinct pickled functions must not silently run the wrong one.
        key = bytes(func_blob)
        func = func_cache.get(key)
        if func is None:
            func = serialization.loads(func_blob)
            func_cache[key] = func

        values, failure = _execute_chunk(func, args, star, kwds)
        if failure is not None and resilient:
            # Resilient pools treat a task exception as a worker fault:
            # deliver nothing, die, and let the master's pending table
            # resubmit the whole chunk (tasks must be idempotent —
            # duplicate deliveries are deduped master-side).
            raise failure[1].rebuild()
        if entry is not None and entry["reduce"] and failure is None:
            # tensor fan-in rides the all-reduce at map end, not the ring
            coll.accumulate(seq, values)
            values = [None] * len(values)
        result = (seq, base, values, failure, ident)
        result_sock.send(serialization.dumps(result), timeout=-1.0)

        tasks_done += len(values)
        if maxtasks is not None and tasks_done >= maxtasks:
            break

    if coll is not None:
        coll.close()
    task_sock.close()
    result_sock.close()


class _WorkerEntry:
    """Picklable worker target carrying resource metadata for the backend."""

    def __init__(self, kwargs, meta):
        self._kwargs = kwargs
        self.__fiber_meta__ = dict(meta or {})

    def __call__(self):
        _pool_worker_loop(**self._kwargs)


# ---------------------------------------------------------------------------
# Master side
# ---------------------------------------------------------------------------


class ZPool:
    """Push/pull worker pool with the multiprocessing.Pool API."""

    resilient = False

    def __init__(
        self,
        processes=None,
        initializer=None,
        initargs=(),
        maxtasksperchild=None,
        gpu_per_worker=None,
        cpu_per_worker=None,
        name=None,
        collective=False,
        collective_backend=None,
    ):
        conf = fam_config.get_object()
        self._processes = processes or os.cpu_count() or 1
        self._nproc_per_job = int(cpu_per_worker or conf.cpu_per_job or 1)
        self._maxtasks = maxtasksperchild
        self._name = name or util.random_name("fam-pool")
        self._gpu_per_worker = gpu_per_worker
        self._meta = {}
        if gpu_per_worker:
            self._meta["gpu"] = gpu_per_worker

        # RCCL data plane: a pool-owned communicator over
        # the workers.  Deterministic membership is required (rank ==
        # worker slot), so it is a ZPool feature; ResilientZPool's
        # anonymous-requeue scheduling is incompatible by design.
        self._collective = bool(collective)
        self._group_master = None
        self._ctl_socks = {}       # ident -> Socket("w") per-worker ctl
        self._ctl_lock = threading.Lock()
        self._slots = [None] * self._processes  # slot -> ident
        self._coll_seqs = set()    # in-flight collective map seqs
        self._coll_dropped = set()  # failed seqs (orphan chunks linger)
        self._reduce_pending = {}  # seq -> spec (reduce_go not yet sent)
        self._reduce_results = {}  # seq -> [Event, tensor, error]
        if self._collective:
            if self.resilient:
                raise ValueError(
                    "collective=True requires ZPool: in-flight "
                    "collectives cannot be resubmitted to a different "
                    "rank (use ZPool and handle map failures yourself)"
                )
            if self._nproc_per_job > 1:
                raise ValueError("collective=True needs one rank per "
                                 "worker process (cpu_per_worker == 1)")
            from .collective import GroupMaster

            # Backend follows the POOL's placement, not the machine:
            # GPU-pinned workers (one device each) ride RCCL; CPU
            # workers use gloo even on a GPU node (unpinned workers
            # would otherwise all claim device 0 — NCCL forbids two
            # ranks per device).
            backend = collective_backend or (
                "nccl" if gpu_per_worker else "gloo"
            )
            self._group_master = GroupMaster(self._processes, backend)

        self._init_blob = None
        if initializer is not None:
            self._init_blob = serialization.dumps_closure(
                (initializer, tuple(initargs))
            )

        mode = "rep" if self.resilient else "w"
        self._task_sock = Socket(mode, self._name + ".task", bind=True)
        self._result_sock = Socket("r", self._name + ".res", bind=True)

        self._inventory = Inventory()
        self._taskq = _stdlib_queue.Queue()
        self._workers = {}  # ident -> Process
        self._callback_watch = {}  # seq -> AsyncResult with callbacks
        self._all_idents = set()
        self._worker_lock = threading.Lock()
        self._state = "run"  # run -> closing -> terminated
        self._workers_started = False
        self._sent = 0
        self._recv = 0
        self._max_inflight = conf.max_inflight
        self._pending = {}  # ident -> {(seq, base): task}  (resilient only)

        self._result_thread = threading.Thread(
            target=self._result_loop, name="fam-pool-results", daemon=True
        )
        self._result_thread.start()
        self._dispatch_thread = threading.Thread(
            target=self._dispatch_loop, name="fam-pool-dispatch", daemon=True
        )
        self._dispatch_thread.start()
        self._worker_thread = None

    # -- worker management -------------------------------------------------
    def _lazy_start_workers(self, func):
        if self._workers_started:
            return
        meta = getattr(func, "__fiber_meta__", None)
        if meta:
            if self._meta and self._meta != dict(meta):
                raise ValueError(
                    "conflicting resource meta: pool=%r func=%r"
                    % (self._meta, meta)
                )
            self._meta = dict(meta)
        self._workers_started = True
        self._worker_thread = threading.Thread(
            target=self._worker_loop, name="fam-pool-workers", daemon=True
        )
        self._worker_thread.start()

    def _spawn_worker(self, slot):
        ident = util.random_name("w")[:24]
        self._all_idents.add(ident)
        kwargs = dict(
            task_addr=self._name + ".task",
            result_addr=self._name + ".res",
            resilient=self.resilient,
            maxtasks=self._maxtasks,
            init_blob=self._init_blob,
            ident_prefix=ident,
            nproc=self._nproc_per_job,
        )
        if self._collective:
            # rank == slot: a respawned worker inherits the dead one's
            # rank, so the communicator shape is stable across deaths.
            ctl_addr = "%s.ctl.%s" % (self._name, ident)
            with self._ctl_lock:
                self._ctl_socks[ident] = Socket("w", ctl_addr, bind=True)
            kwargs["ctl_addr"] = ctl_addr
            desc = self._group_master.descriptor(rank=slot)
            desc["dropped_seqs"] = sorted(self._coll_dropped)
            kwargs["group_desc"] = desc
        entry = _WorkerEntry(kwargs, self._meta)
        proc = Process(
            target=entry, name="%s-worker-%d" % (self._name, slot)
        )
        # Register BEFORE start so the dispatcher can attribute the
        # worker's very first task request.
        with self._worker_lock:
            self._workers[ident] = proc
            self._slots[slot] = ident
        proc.start()
        return ident, proc

    def _worker_loop(self):
        """Maintain the worker population; resubmit a dead worker's tasks."""
        crash_streak = 0
        crash_recv_mark = -1
        while self._state == "run":
            with self._worker_lock:
                dead = [
                    (ident, proc)
                    for ident, proc in self._workers.items()
                    if proc.exitcode is not None
                ]
                for ident, proc in dead:
                    del self._workers[ident]
                    for slot, sid in enumerate(self._slots):
                        if sid == ident:
                            self._slots[slot] = None
            for ident, proc in dead:
                self._on_worker_death(ident, proc)
            if dead and self._collective:
                self._rebuild_group([i for i, _ in dead])
            if dead:
                # Crash-loop detection: workers dying repeatedly while
                # tasks are outstanding and NO results arrive means every
                # respawn hits the same fatal error (e.g. the task
                # function cannot be unpickled worker-side).  Without
                # this, map() hangs forever; with it, outstanding maps
                # fail with the dead worker's own log.
                if self._recv == crash_recv_mark and self._sent > self._recv:
                    crash_streak += len(dead)
                else:
                    crash_streak = len(dead)
                    crash_recv_mark = self._recv
                if crash_streak >= 2 * self._processes + 2:
                    logs = ""
                    try:
                        logs = dead[-1][1].logs()[-1500:]
                    except Exception:  # noqa: BLE001
                        pass
                    self._inventory.fail_all(RuntimeError(
                        "pool workers are crash-looping (%d consecutive "
                        "deaths with zero results while %d tasks are in "
                        "flight); last worker log:\n%s"
                        % (crash_streak, self._sent - self._recv, logs)
                    ))
                    crash_streak = 0
                    crash_recv_mark = self._recv
            with self._worker_lock:
                free = [s for s, sid in enumerate(self._slots)
                        if sid is None]
            for slot in free:
                if self._state != "run":
                    break
                try:
                    self._spawn_worker(slot)
                except Exception:
                    if self._state == "run":
                        util.get_logger().exception("worker spawn failed")
                    time.sleep(0.5)
            time.sleep(0.1)

    def _rebuild_group(self, dead_idents):
        """A member died: the communicator generation is invalid.  Fail
        every in-flight collective map with diagnosis (Ring-failure
        policy: fail fast, rebuild for subsequent work), rotate the
        rendezvous, and tell the survivors."""
        for ident in dead_idents:
            with self._ctl_lock:
                sock = self._ctl_socks.pop(ident, None)
            if sock is not None:
                sock.close()
        exc = RuntimeError(
            "pool worker(s) %s died during a collective map; the "
            "communicator was rebuilt — resubmit the map" % dead_idents
        )
        failed = list(self._coll_seqs)
        for seq in failed:
            self._coll_seqs.discard(seq)
            self._coll_dropped.add(seq)
            self._reduce_pending.pop(seq, None)
            holder = self._reduce_results.get(seq)
            if holder is not None:
                holder[2] = exc
                holder[0].set()
            self._inventory.fail_seq(seq, exc)
        desc = self._group_master.rebuild()
        self._ctl_broadcast(lambda rank, ident: ("rebuild", desc))
        for seq in failed:  # orphan chunks must not wedge survivors
            self._ctl_broadcast(lambda rank, ident: ("drop", seq))
        util.get_logger().warning(
            "collective group rebuilt (gen %d) after death of %s",
            desc["gen"], dead_idents,
        )

    def _ctl_broadcast(self, make_msg):
        """Send a ctl record to every live worker under ONE lock hold —
        the global ordering guarantee the worker-side collective
        sequencing relies on.  make_msg(rank, ident) -> tuple."""
        with self._ctl_lock:
            with self._worker_lock:
                slots = list(enumerate(self._slots))
            for rank, ident in slots:
                if ident is None:
                    continue
                sock = self._ctl_socks.get(ident)
                if sock is None:
                    continue
                try:
                    sock.send(
                        serialization.dumps(make_msg(rank, ident)),
                        timeout=-1.0,
                    )
                except (RuntimeError, OSError):
                    pass  # dying worker; the reaper handles it

    def _on_worker_death(self, ident, proc):
        pass  # resilient subclass resubmits

    # -- data plane --------------------------------------------------------
    def _dispatch_loop(self):
        """Drain the local task queue into the task ring.  Bursts of
        queued chunks go out through one batched ring op (one lock hold
        + one wake); the resilient subclass overrides _send_tasks to
        keep its per-request credit scheduling."""
        import queue as _stdq

        while True:
            task = self._taskq.get()
            if task is None:
                return
            while (
                self._sent - self._recv > self._max_inflight
                and self._state == "run"
            ):
                time.sleep(0.005)
            if self._state == "terminated":
                continue
            batch = [task]
            sentinel = False
            while len(batch) < 32:
                try:
                    nxt = self._taskq.get_nowait()
                except _stdq.Empty:
                    break
                if nxt is None:
                    sentinel = True
                    break
                batch.append(nxt)
            try:
                self._send_tasks(batch)
            except RuntimeError:
                return  # ring closed
            self._sent += len(batch)
            if sentinel:
                return

    def _send_tasks(self, tasks):
        payloads = [serialization.dumps(t) for t in tasks]
        self._task_sock._rings["main"].send_many(payloads, -1.0)

    def _send_task(self, task):
        self._task_sock.send(serialization.dumps(task), timeout=-1.0)

    def _result_loop(self):
        ring = self._result_sock._rings["main"]
        while True:
            try:
                # batched drain: one lock hold + one wake per burst of
                # completed chunks (matters at small chunksizes)
                payloads = ring.recv_many(32, 0.2)
            except RuntimeError:
                return
            except Exception:
                return
            if self._state == "terminated":
                return
            for payload in payloads:
                seq, base, values, failure, ident = (
                    serialization.loads(payload)
                )
                self._recv += 1
                if base == _REDUCE_BASE:
                    # reduced tensor from rank 0, or a worker-side
                    # collective failure report
                    if failure is not None:
                        self._coll_seqs.discard(seq)
                        self._reduce_pending.pop(seq, None)
                        exc = failure[1].rebuild()
                        holder = self._reduce_results.get(seq)
                        if holder is not None and not holder[0].is_set():
                            holder[2] = exc
                            holder[0].set()
                        self._inventory.fail_seq(seq, exc)
                        # a communicator fault of the CURRENT generation
                        # poisons gloo/RCCL state on every rank: rotate
                        # to a virgin namespace so retries converge
                        if failure[0] == self._group_master.gen:
                            self._rebuild_group([])
                        continue
                    holder = self._reduce_results.get(seq)
                    if holder is not None:
                        holder[1] = values[0]
                        holder[0].set()
                    self._coll_seqs.discard(seq)
                    continue
                self._ack(ident, seq, base)
                self._inventory.put(seq, base, values, failure)
                if seq in self._reduce_pending and self._inventory.done(seq):
                    spec = self._reduce_pending.pop(seq)
                    done, error, _results = self._inventory.peek(seq)
                    if error is not None:
                        holder = self._reduce_results.get(seq)
                        if holder is not None:
                            holder[2] = error.rebuild()
                            holder[0].set()
                        self._coll_seqs.discard(seq)
                        self._ctl_broadcast(
                            lambda rank, ident: ("drop", seq)
                        )
                    else:
                        # every chunk accumulated worker-side: fire the
                        # all-reduce fan-in
                        self._ctl_broadcast(
                            lambda rank, ident: ("reduce_go", seq)
                        )
                elif seq in self._coll_seqs and self._inventory.done(seq):
                    self._coll_seqs.discard(seq)
                    # free the workers' staged shared tensors (device
                    # memory!) — without this every shared= map leaks
                    # its broadcast copies in each worker forever
                    self._ctl_broadcast(lambda rank, ident: ("drop", seq))
                watcher = self._callback_watch.get(seq)
                if watcher is not None:
                    done, error, results = self._inventory.peek(seq)
                    if done:
                        self._callback_watch.pop(seq, None)
                        watcher._fire_callbacks(error, results)

    def _ack(self, ident, seq, base):
        pass  # resilient subclass clears pending table

    # -- task submission ---------------------------------------------------
    def _check_running(self):
        if self._state != "run":
            raise ValueError("Pool not running")

    def _submit(self, func, iterable, chunksize, starmap, kwds=None,
                single=False, callback=None, error_callback=None,
                shared=None, reduce=None, reduce_spec=None):
        self._check_running()
        self._lazy_start_workers(func)
        flags = 0
        if shared is not None or reduce is not None:
            if not self._collective:
                raise ValueError(
                    "shared=/reduce= need a Pool(collective=True)"
                )
            if reduce is not None and reduce != "sum":
                raise ValueError("reduce must be 'sum' (or None)")
            if reduce is not None and reduce_spec is None:
                raise ValueError(
                    "reduce='sum' needs reduce_spec=(shape, dtype): a "
                    "worker that drew no chunk must still contribute "
                    "zeros of the right shape to the all-reduce"
                )
            if reduce is not None and (callback or error_callback):
                raise ValueError(
                    "callbacks are not supported with reduce= (use "
                    "ReduceResult.get/wait)"
                )
            flags |= _FLAG_COLLECTIVE
        items = list(iterable)
        n = len(items)
        seq = self._inventory.add(n)
        if reduce is not None:
            result = ReduceResult(self, seq)
        else:
            result = AsyncResult(
                self, seq, n, callback=callback,
                error_callback=error_callback, single=single,
            )
        if flags:
            self._begin_collective(seq, shared, reduce, reduce_spec)
        if n == 0:
            self._inventory.put(seq, 0, [], None)
            if reduce is not None:
                # no chunks will arrive, so fire the fan-in directly:
                # every worker contributes zeros and rank 0 returns the
                # zero tensor
                self._reduce_pending.pop(seq, None)
                self._ctl_broadcast(lambda rank, ident: ("reduce_go", seq))
            if callback is not None:
                result._fire_callbacks(None, [])
            return result
        if callback is not None or error_callback is not None:
            self._callback_watch[seq] = result
        func_blob = serialization.dumps_closure(func)
        if chunksize is None:
            chunksize = max(1, min(DEFAULT_CHUNKSIZE, n // 4 or 1))
        for base in range(0, n, chunksize):
            chunk = items[base : base + chunksize]
            self._taskq.put(
                (seq, base, func_blob, chunk, starmap, kwds, flags)
            )
        return result

    def _wait_all_slots(self, timeout=60.0):
        """Block until every worker slot is spawned (its ctl ring exists
        master-side, so broadcasts sent from now on cannot be lost even
        if the worker process is still booting)."""
        deadline = time.monotonic() + timeout
        while True:
            with self._worker_lock:
                if all(s is not None for s in self._slots):
                    return
            if self._state != "run" or time.monotonic() > deadline:
                raise TimeoutError("pool workers failed to spawn")
            time.sleep(0.01)

    def _begin_collective(self, seq, shared, reduce, reduce_spec):
        """Stage a collective map: ONE blob of the shared tensors (HIP
        IPC handles for device tensors) goes to rank 0; everyone else
        gets shapes/dtypes; the device-to-device broadcast happens
        worker-side over the pool communicator."""
        self._group_master.ensure()
        self._wait_all_slots()
        meta = None
        blob = None
        if shared:
            meta = {
                name: (tuple(t.shape), str(t.dtype))
                for name, t in shared.items()
            }
            blob = serialization.dumps(dict(shared))
        spec = None
        if reduce is not None:
            shape, dtype = reduce_spec
            spec = (tuple(shape), str(dtype))
            self._reduce_pending[seq] = spec
            self._reduce_results[seq] = [threading.Event(), None, None]
        self._coll_seqs.add(seq)
        self._ctl_broadcast(
            lambda rank, ident: (
                "begin", seq, meta, blob if rank == 0 else None,
                reduce, spec,
            )
        )

    def run_on_all(self, func, args=(), kwds=None, timeout=None):
        """SPMD fan-out: run ``func(*args)`` once on EVERY worker (by
        rank), collect one result per rank, ordered by rank.  The
        function runs with :func:`current_worker_group` available, so it
        can drive its own RCCL collectives — this is the ES-through-pool
        hot path (reference analog: pool.map of rollout shards,
        /root/reference/examples/gecco-2020/es.py:17-34, upgraded to
        SPMD + xGMI collectives)."""
        self._check_running()
        if not self._collective:
            raise ValueError("run_on_all needs a Pool(collective=True)")
        self._lazy_start_workers(func)
        self._group_master.ensure()
        self._wait_all_slots()
        seq = self._inventory.add(self._processes)
        self._coll_seqs.add(seq)
        func_blob = serialization.dumps_closure(func)
        self._ctl_broadcast(
            lambda rank, ident: ("exec", seq, func_blob, tuple(args),
                                 kwds)
        )
        result = AsyncResult(self, seq, self._processes)
        try:
            return result.get(timeout)
        finally:
            self._coll_seqs.discard(seq)

    def stats(self):
        """Observability counters (reference only had sent/recv flow
        counters, fiber/pool.py:902-904; extended here)."""
        with self._worker_lock:
            alive = sum(
                1 for p in self._workers.values() if p.exitcode is None
            )
        return {
            "tasks_sent": self._sent,
            "results_received": self._recv,
            "in_flight": self._sent - self._recv,
            "workers_alive": alive,
            "workers_spawned_total": len(self._all_idents),
            "task_ring_depth": self._task_sock.pending,
            "state": self._state,
        }

    # multiprocessing.Pool API --------------------------------------------
    def apply(self, func, args=(), kwds=None):
        return self.apply_async(func, args, kwds).get()

    def apply_async(self, func, args=(), kwds=None, callback=None,
                    error_callback=None):
        return self._submit(
            func, [tuple(args)], 1, True, kwds=kwds or None, single=True,
            callback=callback, error_callback=error_callback,
        )

    def map(self, func, iterable, chunksize=None, shared=None, reduce=None,
            reduce_spec=None):
        return self.map_async(func, iterable, chunksize, shared=shared,
                              reduce=reduce, reduce_spec=reduce_spec).get()

    def map_async(self, func, iterable, chunksize=None, callback=None,
                  error_callback=None, shared=None, reduce=None,
                  reduce_spec=None):
        return self._submit(
            func, iterable, chunksize, False, callback=callback,
            error_callback=error_callback, shared=shared, reduce=reduce,
            reduce_spec=reduce_spec,
        )

    def starmap(self, func, iterable, chunksize=None, shared=None,
                reduce=None, reduce_spec=None):
        return self.starmap_async(
            func, iterable, chunksize, shared=shared, reduce=reduce,
            reduce_spec=reduce_spec,
        ).get()

    def starmap_async(self, func, iterable, chunksize=None, callback=None,
                      error_callback=None, shared=None, reduce=None,
                      reduce_spec=None):
        return self._submit(
            func, iterable, chunksize, True, callback=callback,
            error_callback=error_callback, shared=shared, reduce=reduce,
            reduce_spec=reduce_spec,
        )

    def imap(self, func, iterable, chunksize=1):
        return self._imap_lazy(func, iterable, chunksize, ordered=True)

    def imap_unordered(self, func, iterable, chunksize=1):
        return self._imap_lazy(func, iterable, chunksize, ordered=False)

    def _imap_lazy(self, func, iterable, chunksize, ordered):
        """Streaming imap (stdlib mp.Pool fidelity): a feeder thread pulls
        from the iterator with back-pressure instead of materializing it;
        results are freed as the consumer yields them."""
        self._check_running()
        self._lazy_start_workers(func)
        iterator = iter(iterable)
        seq = self._inventory.add_dynamic(ordered)
        func_blob = serialization.dumps_closure(func)
        window = max(256, 4 * chunksize * self._processes)

        def feeder():
            base = 0
            try:
                while True:
                    if self._state != "run":
                        self._inventory.fail_dynamic(
                            seq, RuntimeError("pool closed during imap")
                        )
                        return
                    if self._inventory.dynamic_backlog(seq, base) > window:
                        time.sleep(0.002)
                        continue
                    chunk = list(itertools.islice(iterator, chunksize))
                    if not chunk:
                        self._inventory.finish_dynamic(seq, base)
                        return
                    self._taskq.put(
                        (seq, base, func_blob, chunk, False, None, 0)
                    )
                    base += len(chunk)
            except Exception as exc:  # noqa: BLE001 (iterator may raise)
                self._inventory.fail_dynamic(seq, exc)

        threading.Thread(
            target=feeder, name="fam-imap-feeder", daemon=True
        ).start()
        return self._inventory.iget_dynamic(seq, ordered)

    # -- lifecycle ---------------------------------------------------------
    def close(self):
        if self._state != "run":
            return
        self._state = "closing"
        # One exit sentinel per worker core rides the normal task channel
        # (via the local queue) so it lands after all real tasks.
        for _ in range(self._processes * self._nproc_per_job):
            self._taskq.put((_SENTINEL_SEQ, 0, b"", [], False, None, 0))

    def terminate(self):
        if self._state == "terminated":
            return
        self._state = "terminated"
        self._taskq.put(None)
        with self._worker_lock:
            for proc in self._workers.values():
                proc.terminate()
        self._inventory.fail_all(RuntimeError("pool terminated"))

    def join(self, timeout=None):
        """Wait for all workers to exit.

        stdlib semantics: after a plain ``close()`` this blocks until the
        workers drain their remaining tasks and exit, however long that
        takes (``timeout=None``).  Pass a number to cap the wait (a pool
        abandoned at the cap is still torn down safely)."""
        if self._state == "run":
            raise ValueError("Pool is still running; call close() or "
                             "terminate() before join()")
        deadline = None if timeout is None else time.monotonic() + timeout
        # Freeze the population first: once the maintainer thread has
        # exited, no new worker can be spawned behind our back.
        if self._worker_thread is not None:
            self._worker_thread.join(timeout=5.0)
        with self._worker_lock:
            workers = list(self._workers.values())
        if self._state == "terminated":
            for proc in workers:
                proc.terminate()  # idempotent; catches post-sweep spawns
        for proc in workers:
            if proc._popen is None:
                continue  # registered but never started (shutdown race)
            if deadline is None:
                proc.join()
            else:
                proc.join(max(0.1, deadline - time.monotonic()))
        with self._worker_lock:
            self._workers.clear()
        if self._state != "terminated":
            self._state = "terminated"
            self._taskq.put(None)
        # Reap service threads BEFORE closing rings so no thread is blocked
        # inside a ring call when the mapping goes away / at interpreter
        # exit (a daemon thread waking into a finalizing interpreter from
        # C++ aborts the process).
        self._dispatch_thread.join(timeout=2.0)
        self._result_thread.join(timeout=2.0)
        if self._worker_thread is not None:
            self._worker_thread.join(timeout=2.0)
        self._task_sock.close()
        self._result_sock.close()
        with self._ctl_lock:
            for sock in self._ctl_socks.values():
                sock.close()
            self._ctl_socks.clear()
        if self._group_master is not None:
            self._group_master.close()
        exc = RuntimeError("pool closed before reduce completed")
        for holder in self._reduce_results.values():
            if not holder[0].is_set():
                holder[2] = exc
                holder[0].set()
        # Workers killed by SIGTERM/SIGKILL never ran their atexit
        # cleanup; reap their per-ident reply rings (resilient mode)
        # master-side, including forked cores' derived idents.
        import glob as _glob

        for ident in self._all_idents:
            names = [ident] + [
                "%s:%d" % (ident, k) for k in range(1, self._nproc_per_job)
            ]
            for name in names:
                base = "/dev/shm/%s.task.r.%s" % (self._name, name)
                try:
                    os.unlink(base)
                except OSError:
                    pass
                # spill segments of the dead worker's reply ring (a
                # SIGKILLed worker never ran its own sweep)
                for path in _glob.glob(base + ".sp.*"):
                    try:
                        os.unlink(path)
                    except OSError:
                        pass

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.terminate()
        self.join()

    def __del__(self):
        try:
            if self._state != "terminated":
                self.terminate()
        except Exception:
            pass


class ResilientZPool(ZPool):
    """REQ/REP pool with pending-table resubmission (reference
    fiber/pool.py:1425-1689)."""

    resilient = True

    def _alive(self, ident):
        primary = ident.split(":", 1)[0]  # derived core idents share a job
        with self._worker_lock:
            proc = self._workers.get(primary)
        return proc is not None and proc.exitcode is None

    def _send_tasks(self, tasks):
        # REQ/REP credit scheduling is inherently per-request: each chunk
        # is handed to a specific live requester and recorded in its
        # pending table, so batching degenerates to a loop.
        for task in tasks:
            self._send_task(task)

    def _send_task(self, task):
        # Serve the next *live* worker request, record attribution, reply.
        # Stale requests from dead workers are discarded (their reply ring
        # has no reader; replying there would strand the task), and a
        # reply that fails because the worker died mid-handoff retracts
        # the task and serves it to the next requester.
        seq = task[0]
        payload = serialization.dumps(task)
        while True:
            request = self._task_sock.recv_request(timeout=0.2)
            if request is None:
                if self._state == "terminated":
                    raise RuntimeError("pool terminated")
                continue
            ident, _ = request
            if not self._alive(ident):
                continue
            if seq != _SENTINEL_SEQ:
                self._pending.setdefault(ident, {})[(seq, task[1])] = task
            try:
                self._task_sock.send_reply(ident, payload)
            except (RuntimeError, OSError):
                # Reply ring closed/unlinked: the worker is gone.
                if seq != _SENTINEL_SEQ:
                    self._pending.get(ident, {}).pop((seq, task[1]), None)
                self._task_sock.drop_peer(ident)
                continue
            if seq != _SENTINEL_SEQ and not self._alive(ident):
                # Died between the liveness check and the reply; the reaper
                # may have already drained its pending table, so pull the
                # entry back ourselves (duplicate delivery is deduped by
                # the Inventory's arrived[] bitmap).
                entry = self._pending.get(ident, {}).pop((seq, task[1]), None)
                if entry is not None:
                    self._taskq.put(entry)
            return

    def _ack(self, ident, seq, base):
        table = self._pending.get(ident)
        if table is not None:
            table.pop((seq, base), None)

    def _on_worker_death(self, ident, proc):
        """Requeue everything the dead job's cores had claimed."""
        dead_idents = [
            k
            for k in list(self._pending)
            if k == ident or k.startswith(ident + ":")
        ] or [ident]
        for did in dead_idents:
            table = self._pending.pop(did, None)
            self._task_sock.drop_peer(did)
            if table:
                util.get_logger().warning(
                    "worker %s died with %d pending chunks; resubmitting",
                    did,
                    len(table),
                )
                for task in table.values():
                    self._taskq.put(task)


Pool = ResilientZPool


def __getattr__(name):  # PEP 562 lazy re-export
    """``fiber_amd.pool.ClassicPool`` — the legacy two-queue pool
    (reference fiber/pool.py:175-641 analog).  Lazy: classic_pool
    imports THIS module for Inventory/_execute_chunk, so an eager
    tail-import here is a circular import in any process that loads
    classic_pool first (e.g. a worker unpickling _ClassicEntry)."""
    if name == "ClassicPool":
        from .classic_pool import ClassicPool

        return ClassicPool
    raise AttributeError("module %r has no attribute %r" % (__name__, name))
