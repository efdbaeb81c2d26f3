"""Pool tests (parity: reference tests/test_pool.py)."""

import random
import time

import pytest

import fiber_amd
from fiber_amd.pool import ResilientZPool, ZPool


def _square(x):
    return x * x


def _add(a, b):
    return a + b


def _random_error(x):
    """Chaos worker (reference random_error_worker, test_pool.py:60-68)."""
    random.seed()
    if random.random() < 0.05:
        raise ValueError("chaos")
    return x * 2


def _checksum_bytes(a):
    return int(a.astype("u8").sum())


def _make_big_array(mb):
    import numpy as np

    return np.full(mb << 20, 7, dtype=np.uint8)


def _slow_identity(x):
    time.sleep(0.01)
    return x


@pytest.fixture
def pool():
    p = ZPool(processes=4)
    yield p
    p.terminate()
    p.join()


@pytest.fixture
def rpool():
    p = ResilientZPool(processes=4)
    yield p
    p.terminate()
    p.join()


class TestZPool:
    def test_map(self, pool):
        assert pool.map(_square, range(100)) == [x * x for x in range(100)]

    def test_map_empty(self, pool):
        assert pool.map(_square, []) == []

    def test_apply(self, pool):
        assert pool.apply(_add, (2, 3)) == 5

    def test_apply_async(self, pool):
        assert pool.apply_async(_add, (2, 3)).get(30) == 5

    def test_starmap(self, pool):
        assert pool.starmap(_add, [(1, 2), (3, 4)]) == [3, 7]

    def test_starmap_async(self, pool):
        assert pool.starmap_async(_add, [(1, 2), (3, 4)]).get(30) == [3, 7]

    def test_imap_ordered(self, pool):
        assert list(pool.imap(_square, range(50))) == [
            x * x for x in range(50)
        ]

    def test_imap_unordered(self, pool):
        got = sorted(pool.imap_unordered(_square, range(50)))
        assert got == sorted(x * x for x in range(50))

    def test_map_async_callback(self, pool):
        seen = []
        r = pool.map_async(_square, range(10), callback=seen.append)
        r.get(30)
        assert seen and seen[0] == [x * x for x in range(10)]

    def test_large_map(self, pool):
        n = 5000
        assert pool.map(_square, range(n)) == [x * x for x in range(n)]

    def test_exception_propagates(self, pool):
        def boom(x):
            raise RuntimeError("boom %d" % x)

        with pytest.raises(RuntimeError):
            pool.map(boom, range(8))

    def test_map_after_close_raises(self):
        p = ZPool(processes=2)
        p.map(_square, range(4))
        p.close()
        with pytest.raises(ValueError):
            p.map(_square, range(4))
        p.join()

    def test_uneven_task_durations_balanced(self, pool):
        # demand-driven dispatch: a slow task must not serialize the rest
        t0 = time.monotonic()
        res = pool.map(_slow_identity, range(80), chunksize=1)
        elapsed = time.monotonic() - t0
        assert res == list(range(80))
        # 80 tasks x 10ms / 4 workers = 0.2s ideal; allow generous slack
        assert elapsed < 2.0

    def test_context_manager(self):
        with ZPool(processes=2) as p:
            assert p.map(_square, range(10)) == [x * x for x in range(10)]

    def test_large_payloads_roundtrip(self):
        """Task args/results far past the ring capacity (8 MB default)
        ride the spill path: a 64 MB array round-trips unchanged."""
        import numpy as np

        with ZPool(processes=2) as p:
            arrs = [np.full(16 << 20, i, dtype=np.uint8) for i in range(3)]
            out = p.map(_checksum_bytes, arrs, chunksize=1)
            assert out == [int(a[0]) * len(a) for a in arrs]
            big = p.apply(_make_big_array, (64,))
            assert big.nbytes == 64 << 20
            assert big[0] == 7 and big[-1] == 7


class TestResilientZPool:
    def test_chaos_map_completes(self, rpool):
        res = rpool.map(_random_error, range(300), chunksize=4)
        assert res == [x * 2 for x in range(300)]

    def test_chaos_unordered(self, rpool):
        got = sorted(rpool.imap_unordered(_random_error, range(100),
                                          chunksize=4))
        assert got == sorted(x * 2 for x in range(100))

    def test_worker_kill_mid_map_recovers(self, rpool):
        result = rpool.map_async(_slow_identity, range(200), chunksize=2)
        time.sleep(0.3)
        # murder one live worker mid-flight
        with rpool._worker_lock:
            victim = next(iter(rpool._workers.values()))
        victim.kill()
        assert result.get(60) == list(range(200))

    def test_plain_map(self, rpool):
        assert rpool.map(_square, range(64)) == [x * x for x in range(64)]

    def test_facade_error_handling_flag(self):
        p = fiber_amd.Pool(2, error_handling=True)
        assert isinstance(p, ResilientZPool)
        p.terminate()
        p.join()
        p2 = fiber_amd.Pool(2)
        assert isinstance(p2, ZPool) and not isinstance(p2, ResilientZPool)
        p2.terminate()
        p2.join()


def _pid_task(x):
    import os

    time.sleep(0.05)
    return os.getpid()


class TestMultiWorkerPerJob:
    def test_cpu_per_worker_forks_cores(self):
        """cpu_per_job analog (reference pool.py:861-878): one job hosts
        several worker cores sharing the rings."""
        pool = ZPool(processes=2, cpu_per_worker=2)
        try:
            pids = set(pool.map(_pid_task, range(32), chunksize=1))
            assert len(pids) >= 3  # 2 jobs x 2 cores, allow one laggard
        finally:
            pool.terminate()
            pool.join()

    def test_resilient_with_forked_cores(self):
        pool = ResilientZPool(processes=2, cpu_per_worker=2)
        try:
            res = pool.map(_random_error, range(120), chunksize=2)
            assert res == [x * 2 for x in range(120)]
        finally:
            pool.terminate()
            pool.join()


class TestPoolInitializer:
    def test_initializer_runs_in_workers(self):
        def init(v):
            import os

            os.environ["FAM_TEST_INIT"] = str(v)

        def read_init(_):
            import os

            return os.environ.get("FAM_TEST_INIT")

        p = ZPool(processes=2, initializer=init, initargs=(7,))
        try:
            assert p.map(read_init, range(4)) == ["7"] * 4
        finally:
            p.terminate()
            p.join()


def _slow_square(x):
    time.sleep(0.001)
    return x * x


class TestLazyImap:
    def test_imap_streams_from_generator(self):
        """imap must not materialize the iterable (stdlib fidelity):
        feed a generator that tracks its own progress."""
        pool = ZPool(processes=2)
        pulled = []

        def gen():
            for i in range(5000):
                pulled.append(i)
                yield i

        try:
            it = pool.imap(_square, gen(), chunksize=4)
            first = [next(it) for _ in range(8)]
            assert first == [x * x for x in range(8)]
            # back-pressure: the feeder must NOT have drained all 5000
            assert len(pulled) < 5000
            rest = list(it)
            assert len(rest) == 5000 - 8
            assert rest[-1] == 4999 * 4999
            assert len(pulled) == 5000
        finally:
            pool.terminate()
            pool.join()

    def test_imap_unordered_streams(self):
        pool = ZPool(processes=4)
        try:
            got = sorted(pool.imap_unordered(_slow_square, iter(range(200)),
                                             chunksize=2))
            assert got == sorted(x * x for x in range(200))
        finally:
            pool.terminate()
            pool.join()

    def test_imap_propagates_iterator_error(self):
        pool = ZPool(processes=2)

        def bad_gen():
            yield 1
            raise RuntimeError("iterator exploded")

        try:
            it = pool.imap(_square, bad_gen())
            with pytest.raises(RuntimeError):
                list(it)
        finally:
            pool.terminate()
            pool.join()

    def test_imap_task_error_at_position(self):
        pool = ZPool(processes=2)

        def maybe_boom(x):
            if x == 7:
                raise ValueError("x7")
            return x

        try:
            it = pool.imap(maybe_boom, range(20), chunksize=3)
            got = [next(it) for _ in range(7)]
            assert got == list(range(7))
            with pytest.raises(ValueError):
                next(it)
        finally:
            pool.terminate()
            pool.join()

    def test_resilient_imap_chaos(self):
        pool = ResilientZPool(processes=4)
        try:
            got = sorted(pool.imap_unordered(_random_error, iter(range(150)),
                                             chunksize=2))
            assert got == sorted(x * 2 for x in range(150))
        finally:
            pool.terminate()
            pool.join()


def _inner_times10(x):
    return x * 10


def _nested_outer(n):
    pool = ZPool(processes=2)
    try:
        return sum(pool.map(_inner_times10, range(n)))
    finally:
        pool.terminate()
        pool.join()


class TestNestedPools:
    def test_worker_creates_its_own_pool(self):
        """Nested fiber processes (reference-supported pattern): a pool
        worker spins up its own pool."""
        pool = ZPool(processes=2)
        try:
            assert pool.map(_nested_outer, [3, 4]) == [30, 60]
        finally:
            pool.terminate()
            pool.join()


class TestAsyncCallbacks:
    def test_callback_fires_without_get(self):
        """mp.Pool semantics: callbacks fire from the pool when the job
        completes, independent of .get()."""
        pool = ZPool(processes=2)
        seen = []
        try:
            pool.map_async(_square, range(10), callback=seen.append)
            deadline = time.monotonic() + 30
            while not seen and time.monotonic() < deadline:
                time.sleep(0.01)
            assert seen == [[x * x for x in range(10)]]
        finally:
            pool.terminate()
            pool.join()

    def test_error_callback_fires(self):
        pool = ZPool(processes=2)
        errors = []

        def boom(x):
            raise ValueError("cb-%d" % x)

        try:
            r = pool.map_async(boom, range(4), chunksize=1,
                               error_callback=errors.append)
            deadline = time.monotonic() + 30
            while not errors and time.monotonic() < deadline:
                time.sleep(0.01)
            assert errors and isinstance(errors[0], ValueError)
            with pytest.raises(ValueError):
                r.get(10)
        finally:
            pool.terminate()
            pool.join()

    def test_apply_async_callback(self):
        pool = ZPool(processes=2)
        seen = []
        try:
            pool.apply_async(_add, (2, 3), callback=seen.append)
            deadline = time.monotonic() + 30
            while not seen and time.monotonic() < deadline:
                time.sleep(0.01)
            assert seen == [5]
        finally:
            pool.terminate()
            pool.join()


def _explode_on_load():
    raise RuntimeError("boom on unpickle")


class _ExplodingCallable:
    """Pickles fine master-side; detonates when the worker unpickles it."""

    def __reduce__(self):
        return (_explode_on_load, ())

    def __call__(self, x):  # pragma: no cover - never reached
        return x


class TestCrashLoopDetection:
    def test_unpicklable_function_fails_instead_of_hanging(self):
        """A function that cannot be unpickled worker-side kills every
        respawned worker; the pool must diagnose the crash loop and fail
        the map (with the worker's own log) instead of hanging forever
        (found the expensive way: a heredoc-__main__ function hung a
        GPU lease for its full timeout)."""
        pool = ZPool(processes=2)
        try:
            with pytest.raises(RuntimeError, match="crash-looping"):
                pool.map(_ExplodingCallable(), range(8), chunksize=1)
        finally:
            pool.terminate()
            pool.join()


class TestDelayedBackend:
    def test_pool_correct_with_slow_job_creation(self, monkeypatch):
        """Reference DelayedBackend idiom (uber/fiber
        tests/test_docker_backend.py:86-105): randomized create_job
        delays must not affect pool correctness or ordering."""
        import random as _random

        from fiber_amd import backend as fam_backend
        from fiber_amd.backends.local import Backend as LocalBackend

        class DelayedBackend(LocalBackend):
            def create_job(self, spec):
                time.sleep(_random.uniform(0.0, 0.4))
                return super().create_job(spec)

        monkeypatch.setitem(fam_backend._backends, "local",
                            DelayedBackend())
        pool = ZPool(processes=4)
        try:
            assert pool.map(_square, range(100), chunksize=4) == [
                x * x for x in range(100)
            ]
        finally:
            pool.terminate()
            pool.join()
