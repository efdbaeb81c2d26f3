"""Process lifecycle tests (parity: reference tests/test_process.py)."""

import select
import time

import pytest

import fiber_amd
from fiber_amd.queues import SimpleQueue


def _noop():
    pass


def _fail():
    raise ValueError("expected failure")


def _sleep_forever():
    time.sleep(600)


def _echo_name(q):
    q.put(fiber_amd.current_process().name)


def _exit_code_7():
    import sys

    sys.exit(7)


class TestProcess:
    def test_start_join_exitcode(self):
        p = fiber_amd.Process(target=_noop)
        p.start()
        p.join(30)
        assert p.exitcode == 0

    def test_error_exit(self):
        p = fiber_amd.Process(target=_fail)
        p.start()
        p.join(30)
        assert p.exitcode == 1

    def test_sys_exit_code(self):
        p = fiber_amd.Process(target=_exit_code_7)
        p.start()
        p.join(30)
        assert p.exitcode == 7

    def test_is_alive(self):
        p = fiber_amd.Process(target=_sleep_forever)
        assert not p.is_alive()
        p.start()
        assert p.is_alive()
        p.terminate()
        p.join(30)
        assert not p.is_alive()

    def test_terminate_exitcode(self):
        p = fiber_amd.Process(target=_sleep_forever)
        p.start()
        time.sleep(0.2)
        p.terminate()
        p.join(30)
        assert p.exitcode not in (0, None)

    def test_sentinel_selectable(self):
        p = fiber_amd.Process(target=_noop)
        p.start()
        readable, _, _ = select.select([p.sentinel], [], [], 30)
        assert readable
        p.join(30)
        assert p.exitcode == 0

    def test_active_children(self):
        p = fiber_amd.Process(target=_sleep_forever)
        p.start()
        assert p in fiber_amd.active_children()
        p.terminate()
        p.join(30)
        assert p not in fiber_amd.active_children()

    def test_current_process_name_in_child(self):
        q = SimpleQueue()
        p = fiber_amd.Process(target=_echo_name, args=(q,), name="zed")
        p.start()
        assert q.get(timeout=30) == "zed"
        p.join(30)

    def test_double_start_raises(self):
        p = fiber_amd.Process(target=_noop)
        p.start()
        with pytest.raises(RuntimeError):
            p.start()
        p.join(30)

    def test_process_pickles_without_popen(self):
        import pickle

        p = fiber_amd.Process(target=_noop, name="pk")
        p.start()
        p2 = pickle.loads(pickle.dumps(p))
        assert p2.name == "pk"
        assert p2._popen is None
        p.join(30)


class TestManyFds:
    def test_spawn_with_1100_open_fds(self):
        """Reference regression (tests/test_popen.py:95-113): process
        management must not depend on select()-able fd numbers < 1024."""
        import os

        fds = [os.open("/dev/null", os.O_RDONLY) for _ in range(1100)]
        try:
            p = fiber_amd.Process(target=_noop)
            p.start()
            p.join(60)
            assert p.exitcode == 0
        finally:
            for fd in fds:
                os.close(fd)


class TestStartFailure:
    def test_backend_start_timeout_surfaces(self, monkeypatch):
        """Fault injection at the backend seam (reference TimeoutBackend
        idiom, tests/test_process.py:27-39)."""
        from fiber_amd import backend as fam_backend
        from fiber_amd.backends.local import Backend as LocalBackend

        class FlakyBackend(LocalBackend):
            calls = 0

            def create_job(self, spec):
                FlakyBackend.calls += 1
                if FlakyBackend.calls == 1:
                    raise OSError("injected create_job failure")
                return super().create_job(spec)

        flaky = FlakyBackend()
        monkeypatch.setitem(fam_backend._backends, "local", flaky)
        p = fiber_amd.Process(target=_noop)
        with pytest.raises(OSError):
            p.start()
        # second attempt (fresh Process) succeeds
        p2 = fiber_amd.Process(target=_noop)
        p2.start()
        p2.join(30)
        assert p2.exitcode == 0
