"""ClassicPool — the legacy two-SimpleQueue pool architecture.

Reference parity: uber/fiber ``ClassicPool`` (fiber/pool.py:175-641) is
the original design — tasks ride one shared queue, results another, and
master-side handler threads glue them together.  The reference itself
superseded it with ZPool (its context never exports ClassicPool), and so
does this repo; it exists for API parity and as the canonical example of
building a pool purely from the public queue primitives.

Differences from ZPool (deliberate, matching the reference's split):
* transport is two :class:`fiber_amd.queues.SimpleQueue` objects — the
  queue ends travel to workers as pickled task args, exercising the
  re-dial path;
* no REQ/REP attribution: a dead worker's in-flight chunk is LOST
  (reference ClassicPool semantics — use ResilientZPool for recovery);
  dead workers are still respawned so the pool keeps serving.
"""

import itertools
import threading
import time

from . import serialization, util
from .pool import (
    AsyncResult,
    DEFAULT_CHUNKSIZE,
    Inventory,
    _execute_chunk,
    _ExcInfo,
)
from .process import Process
from .queues import SimpleQueue


def _classic_worker(taskq, resq, init_blob, maxtasks):
    if init_blob is not None:
        initializer, initargs = serialization.loads(init_blob)
        initializer(*initargs)
    func_cache = {}
    done = 0
    while True:
        task = taskq.get()
        if task is None:
            break
        seq, base, func_blob, args, star, kwds = task
        func = func_cache.get(func_blob)
        if func is None:
            func = serialization.loads(func_blob)
            func_cache[func_blob] = func
        values, failure = _execute_chunk(func, args, star, kwds)
        resq.put((seq, base, values, failure))
        done += len(values)
        if maxtasks is not None and done >= maxtasks:
            break


class _ClassicEntry:
    def __init__(self, taskq, resq, init_blob, maxtasks):
        self._args = (taskq, resq, init_blob, maxtasks)

    def __call__(self):
        _classic_worker(*self._args)


class ClassicPool:
    def __init__(self, processes=None, initializer=None, initargs=(),
                 maxtasksperchild=None):
        import os

        self._processes = processes or os.cpu_count() or 1
        self._maxtasks = maxtasksperchild
        self._taskq = SimpleQueue()
        self._resq = SimpleQueue()
        self._inventory = Inventory()
        self._state = "run"
        self._lock = threading.Lock()
        self._workers = []
        self._init_blob = None
        if initializer is not None:
            self._init_blob = serialization.dumps_closure(
                (initializer, tuple(initargs))
            )
        self._entry = _ClassicEntry(self._taskq, self._resq,
                                    self._init_blob, self._maxtasks)
        for i in range(self._processes):
            self._spawn(i)
        self._result_thread = threading.Thread(
            target=self._result_loop, name="fam-classic-results",
            daemon=True,
        )
        self._result_thread.start()
        self._maintainer = threading.Thread(
            target=self._maintain, name="fam-classic-workers", daemon=True
        )
        self._maintainer.start()

    def _spawn(self, index):
        proc = Process(target=self._entry,
                       name="fam-classic-worker-%d" % index)
        proc.start()
        with self._lock:
            self._workers.append(proc)

    def _maintain(self):
        """Respawn dead workers (their in-flight chunk is lost — classic
        semantics; reference fiber/pool.py worker_handler analog)."""
        index = itertools.count(self._processes)
        while self._state == "run":
            with self._lock:
                dead = [p for p in self._workers if p.exitcode is not None]
                for p in dead:
                    self._workers.remove(p)
            for _ in dead:
                if self._state != "run":
                    break
                try:
                    self._spawn(next(index))
                except Exception:  # noqa: BLE001
                    util.get_logger().exception("classic respawn failed")
                    time.sleep(0.5)
            time.sleep(0.2)

    def _result_loop(self):
        while True:
            try:
                records = self._resq.get_many(max_n=64, timeout=0.2)
            except Exception:  # noqa: BLE001 (queue closed)
                return
            if self._state == "terminated":
                return
            for seq, base, values, failure in records:
                self._inventory.put(seq, base, values, failure)

    # -- submission --------------------------------------------------------
    def _check(self):
        if self._state != "run":
            raise ValueError("Pool not running")

    def _submit(self, func, iterable, chunksize, star, kwds=None,
                single=False):
        self._check()
        items = list(iterable)
        n = len(items)
        seq = self._inventory.add(n)
        result = AsyncResult(self, seq, n, single=single)
        if n == 0:
            self._inventory.put(seq, 0, [], None)
            return result
        blob = serialization.dumps_closure(func)
        if chunksize is None:
            chunksize = max(1, min(DEFAULT_CHUNKSIZE, n // 4 or 1))
        for base in range(0, n, chunksize):
            self._taskq.put(
                (seq, base, blob, items[base:base + chunksize], star, kwds)
            )
        return result

    def apply(self, func, args=(), kwds=None):
        return self._submit(func, [tuple(args)], 1, True, kwds=kwds,
                            single=True).get()

    def apply_async(self, func, args=(), kwds=None):
        return self._submit(func, [tuple(args)], 1, True, kwds=kwds,
                            single=True)

    def map(self, func, iterable, chunksize=None):
        return self._submit(func, iterable, chunksize, False).get()

    def map_async(self, func, iterable, chunksize=None):
        return self._submit(func, iterable, chunksize, False)

    def starmap(self, func, iterable, chunksize=None):
        return self._submit(func, iterable, chunksize, True).get()

    def imap(self, func, iterable, chunksize=1):
        result = self._submit(func, iterable, chunksize, False)
        return self._inventory.iget_ordered(result._seq)

    def imap_unordered(self, func, iterable, chunksize=1):
        result = self._submit(func, iterable, chunksize, False)
        return self._inventory.iget_unordered(result._seq)

    # -- lifecycle ---------------------------------------------------------
    def close(self):
        if self._state != "run":
            return
        self._state = "closing"
        for _ in range(self._processes):
            self._taskq.put(None)

    def terminate(self):
        if self._state == "terminated":
            return
        self._state = "terminated"
        with self._lock:
            for proc in self._workers:
                proc.terminate()
        self._inventory.fail_all(RuntimeError("pool terminated"))

    def join(self, timeout=None):
        if self._state == "run":
            raise ValueError("call close() or terminate() before join()")
        if self._maintainer.is_alive():
            self._maintainer.join(timeout=5.0)
        deadline = None if timeout is None else time.monotonic() + timeout
        with self._lock:
            workers = list(self._workers)
        for proc in workers:
            if deadline is None:
                proc.join()
            else:
                proc.join(max(0.1, deadline - time.monotonic()))
        if self._state != "terminated":
            self._state = "terminated"
        self._result_thread.join(timeout=2.0)
        self._taskq.close()
        self._resq.close()

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.terminate()
        self.join()
