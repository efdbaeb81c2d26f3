"""@meta resource-hint tests (parity: reference fiber/meta.py semantics)."""

import pytest

import fiber_amd
from fiber_amd.meta import get_meta, meta


class TestMeta:
    def test_attaches_metadata(self):
        @meta(cpu=2, gpu=1)
        def f():
            return 1

        assert f.__fiber_meta__ == {"cpu": 2, "gpu": 1}
        assert get_meta(f) == {"cpu": 2, "gpu": 1}
        assert f() == 1

    def test_invalid_key_raises(self):
        with pytest.raises(ValueError):
            meta(disk=100)

    def test_facade_export(self):
        assert fiber_amd.meta is meta

    def test_pool_meta_conflict_raises(self):
        """Pool-level and func-level resource hints must agree (reference
        fiber/pool.py:1122-1137 conflict error)."""
        from fiber_amd.pool import ZPool

        @meta(gpu=2)
        def g(x):
            return x

        pool = ZPool(processes=1, gpu_per_worker=1)
        try:
            with pytest.raises(ValueError):
                pool.map(g, [1])
        finally:
            pool.terminate()
            pool.join()
