"""Manager tests (parity: reference tests/test_managers.py)."""

import time

import pytest

import fiber_amd
from fiber_amd.managers import AsyncManager, SyncManager
from fiber_amd.queues import SimpleQueue


class Counter:
    def __init__(self):
        self.n = 0

    def add(self, k=1):
        self.n += k
        return self.n

    def slow(self, delay):
        time.sleep(delay)
        return delay


def _child_mutates(d, out):
    d["child"] = 1
    out.put(d["x"])


@pytest.fixture
def manager():
    m = SyncManager()
    m.start()
    yield m
    m.shutdown()


class TestSyncManager:
    def test_dict(self, manager):
        d = manager.dict()
        d["x"] = 5
        assert d["x"] == 5
        assert len(d) == 1
        assert "x" in d
        del d["x"]
        assert "x" not in d

    def test_list(self, manager):
        l = manager.list()
        l.append(1)
        l.extend([2, 3])
        assert list(l) == [1, 2, 3]
        assert l[1] == 2
        l[1] = 20
        assert list(l) == [1, 20, 3]
        assert len(l) == 3

    def test_namespace(self, manager):
        ns = manager.Namespace()
        ns.alpha = "a"
        assert ns.alpha == "a"

    def test_value_array(self, manager):
        v = manager.Value("i", 3)
        assert v.value == 3
        v.value = 9
        assert v.value == 9
        a = manager.Array("d", [1.0, 2.0])
        assert a[0] == 1.0
        a[0] = 5.0
        assert a[0] == 5.0

    def test_queue(self, manager):
        q = manager.Queue()
        q.put("m")
        assert q.get() == "m"

    def test_nested_managed_objects(self, manager):
        """Nested mutation semantics (reference test_managers.py:65-90)."""
        d = manager.dict()
        inner = manager.list()
        d["inner"] = inner
        d["inner"].append(42)
        assert list(inner) == [42]

    def test_proxy_travels_to_child(self, manager):
        d = manager.dict()
        d["x"] = 10
        out = SimpleQueue()
        p = fiber_amd.Process(target=_child_mutates, args=(d, out))
        p.start()
        assert out.get(timeout=30) == 10
        p.join(30)
        assert p.exitcode == 0
        assert d["child"] == 1
        out.close()

    def test_unsupported_types_not_registered(self, manager):
        # Lock/Semaphore/Event are not supported (reference parity:
        # fiber/managers.py:624-633 comments them out).
        with pytest.raises(AttributeError):
            manager.Lock()

    def test_exception_propagates(self, manager):
        d = manager.dict()
        with pytest.raises(KeyError):
            d["missing"]


class TestAsyncManager:
    def test_async_handle(self):
        AsyncManager.register("Counter", Counter)
        m = AsyncManager()
        m.start()
        try:
            c = m.Counter()
            h1 = c.add(5)
            h2 = c.add(2)
            assert h1.get() == 5
            assert h2.get() == 7
        finally:
            m.shutdown()

    def test_four_managers_parallel(self):
        """4 managers x 1s calls must finish < 2s — proves the calls run
        concurrently (reference test_managers.py:92-119)."""
        AsyncManager.register("Counter", Counter)
        managers = []
        try:
            for _ in range(4):
                m = AsyncManager()
                m.start()
                managers.append(m)
            counters = [m.Counter() for m in managers]
            t0 = time.monotonic()
            handles = [c.slow(1.0) for c in counters]
            assert [h.get() for h in handles] == [1.0] * 4
            assert time.monotonic() - t0 < 2.0
        finally:
            for m in managers:
                m.shutdown()
