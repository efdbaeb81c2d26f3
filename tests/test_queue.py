"""Pipe / SimpleQueue tests (parity: reference tests/test_queue.py)."""

import queue as stdlib_queue
import time

import pytest

import fiber_amd
from fiber_amd.queues import Pipe, SimpleQueue


def _echo(conn, n):
    for _ in range(n):
        conn.send(conn.recv())


def _pump(q_in, q_out, n):
    for _ in range(n):
        q_out.put(q_in.get())


def _consume(q, out, n):
    got = 0
    for _ in range(n):
        q.get()
        got += 1
    out.put(got)


class TestPipe:
    def test_duplex_same_process(self):
        a, b = Pipe()
        a.send({"k": [1, 2]})
        assert b.recv() == {"k": [1, 2]}
        b.send("back")
        assert a.recv() == "back"
        a.close()
        b.close()

    def test_simplex(self):
        reader, writer = Pipe(duplex=False)
        writer.send(123)
        assert reader.recv() == 123
        reader.close()
        writer.close()

    def test_across_processes(self):
        a, b = Pipe()
        p = fiber_amd.Process(target=_echo, args=(b, 3))
        p.start()
        for i in range(3):
            a.send(i * 10)
            assert a.recv(timeout=30) == i * 10
        p.join(30)
        assert p.exitcode == 0
        a.close()

    def test_poll(self):
        a, b = Pipe()
        assert not a.poll(0.05)
        b.send(1)
        assert a.poll(1.0)
        assert a.recv() == 1
        a.close()
        b.close()

    def test_pickled_connection_redials(self):
        import pickle

        a, b = Pipe()
        b2 = pickle.loads(pickle.dumps(b))
        a.send("x")
        assert b2.recv(timeout=5) == "x"
        a.close()
        b.close()


class TestSimpleQueue:
    def test_put_get(self):
        q = SimpleQueue()
        q.put(1)
        q.put("two")
        assert q.get() == 1
        assert q.get() == "two"
        q.close()

    def test_empty_qsize(self):
        q = SimpleQueue()
        assert q.empty()
        q.put(0)
        assert q.qsize() == 1
        q.get()
        assert q.empty()
        q.close()

    def test_get_nowait_empty(self):
        q = SimpleQueue()
        with pytest.raises(stdlib_queue.Empty):
            q.get_nowait()
        q.close()

    def test_through_processes(self):
        q_in, q_out = SimpleQueue(), SimpleQueue()
        p = fiber_amd.Process(target=_pump, args=(q_in, q_out, 10))
        p.start()
        for i in range(10):
            q_in.put(i)
        got = [q_out.get(timeout=30) for _ in range(10)]
        assert got == list(range(10))
        p.join(30)
        q_in.close()
        q_out.close()

    def test_fair_consumption_four_workers(self):
        """Fairness analog of reference test_queue.py:218-250: 4 workers x
        600 msgs each; every worker must receive its full share."""
        q = SimpleQueue()
        out = SimpleQueue()
        workers = [
            fiber_amd.Process(target=_consume, args=(q, out, 600))
            for _ in range(4)
        ]
        for w in workers:
            w.start()
        for i in range(2400):
            q.put(i)
        counts = [out.get(timeout=60) for _ in range(4)]
        assert counts == [600] * 4
        for w in workers:
            w.join(30)
            assert w.exitcode == 0
        q.close()
        out.close()

    def test_large_messages(self):
        q = SimpleQueue()
        blob = b"z" * (1 << 20)
        q.put(blob)
        assert q.get() == blob
        q.close()

    def test_timeout(self):
        q = SimpleQueue()
        t0 = time.monotonic()
        with pytest.raises(TimeoutError):
            q.get(timeout=0.2)
        assert time.monotonic() - t0 < 2.0
        q.close()


class TestSimpleQueueBatched:
    def test_put_many_get_many(self):
        q = SimpleQueue()
        try:
            q.put_many([{"i": i} for i in range(10)])
            got = q.get_many(max_n=6, timeout=5.0)
            assert got == [{"i": i} for i in range(6)]
            got += q.get_many(max_n=64, timeout=5.0)
            assert got == [{"i": i} for i in range(10)]
            assert q.get_many(max_n=4, timeout=0.0) == []
        finally:
            q.close()

    def test_batched_across_processes(self):
        import functools

        q = SimpleQueue()
        out = SimpleQueue()
        try:
            p = fiber_amd.Process(
                target=functools.partial(_batch_echo, q, out), name="qb"
            )
            p.start()
            q.put_many(list(range(100)))
            got = []
            while len(got) < 100:
                items = out.get_many(max_n=64, timeout=30.0)
                assert items, "starved at %d" % len(got)
                got.extend(items)
            assert got == [i * 2 for i in range(100)]
            p.join(30)
            assert p.exitcode == 0
        finally:
            q.close()
            out.close()


def _batch_echo(q, out):
    done = 0
    while done < 100:
        items = q.get_many(max_n=32, timeout=30.0)
        out.put_many([i * 2 for i in items])
        done += len(items)
