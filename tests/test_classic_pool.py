"""ClassicPool — the legacy two-SimpleQueue architecture (reference
fiber/pool.py:175-641 parity; superseded by ZPool there and here)."""

import time

import pytest

from fiber_amd.pool import ClassicPool


def _square(x):
    return x * x


def _add(a, b):
    return a + b


def _slowish(x):
    time.sleep(0.005)
    return x


class TestClassicPool:
    def test_map_apply_starmap(self):
        with ClassicPool(processes=3) as p:
            assert p.map(_square, range(20)) == [x * x for x in range(20)]
            assert p.apply(_add, (3, 4)) == 7
            assert p.starmap(_add, [(1, 2), (3, 4)]) == [3, 7]

    def test_imap_ordered_and_unordered(self):
        with ClassicPool(processes=2) as p:
            assert list(p.imap(_square, range(12), chunksize=2)) == [
                x * x for x in range(12)
            ]
            got = sorted(p.imap_unordered(_square, range(12), chunksize=3))
            assert got == sorted(x * x for x in range(12))

    def test_close_drains_then_join(self):
        p = ClassicPool(processes=2)
        res = p.map_async(_slowish, range(40), chunksize=2)
        p.close()
        assert res.get(60) == list(range(40))
        p.join()

    def test_exception_propagates(self):
        def boom(x):
            raise RuntimeError("classic boom")

        with ClassicPool(processes=2) as p:
            with pytest.raises(RuntimeError):
                p.map(boom, range(4))

    def test_respawn_keeps_serving(self):
        p = ClassicPool(processes=2)
        try:
            assert p.map(_square, range(8)) == [x * x for x in range(8)]
            with p._lock:
                victim = p._workers[0]
            victim.kill()
            time.sleep(0.6)  # maintainer respawns
            assert p.map(_square, range(8)) == [x * x for x in range(8)]
        finally:
            p.terminate()
            p.join()
