"""Misc parity tests (reference tests/test_misc.py, test_popen.py,
test_docker_backend.py analogs)."""

import threading
import time

import fiber_amd
from fiber_amd import config as fam_config
from fiber_amd import util
from fiber_amd.pool import ZPool


def _noop():
    pass


def _sleepy(t):
    time.sleep(t)
    return t


def _log_something():
    log = util.get_logger()
    log.warning("CHILD-LINE-MARKER")


class TestSingleAdminThread:
    def test_one_accept_thread_for_many_processes(self):
        """Reference test_popen.py:69-93: exactly one background admin
        thread multiplexes all children."""
        procs = [fiber_amd.Process(target=_noop) for _ in range(4)]
        for p in procs:
            p.start()
        admin_threads = [
            t for t in threading.enumerate() if t.name == "fam_admin"
        ]
        assert len(admin_threads) == 1
        for p in procs:
            p.join(30)
            assert p.exitcode == 0


class TestDelayedBackend:
    def test_pool_correct_under_slow_job_creation(self, monkeypatch):
        """Reference DelayedBackend (test_docker_backend.py:86-105):
        create_job delays must not affect pool correctness."""
        import random

        from fiber_amd import backend as fam_backend
        from fiber_amd.backends.local import Backend as LocalBackend

        class DelayedBackend(LocalBackend):
            def create_job(self, spec):
                time.sleep(random.uniform(0.05, 0.3))
                return super().create_job(spec)

        monkeypatch.setitem(fam_backend._backends, "local", DelayedBackend())
        pool = ZPool(processes=3)
        try:
            assert pool.map(_sleepy, [0.01] * 30) == [0.01] * 30
        finally:
            pool.terminate()
            pool.join()


class TestLogSeparation:
    def test_master_log_does_not_contain_child_lines(self, tmp_path):
        """Reference test_misc.py:182-221: per-process log files."""
        log_base = str(tmp_path / "fam.log")
        fam_config.init(log_file=log_base, log_level="info")
        util.init_logger(fam_config.get_object(), "MainProcess")
        try:
            proc = fiber_amd.Process(target=_log_something, name="logchild")
            proc.start()
            proc.join(30)
            assert proc.exitcode == 0
            child_log = tmp_path / "fam.log.logchild"
            assert child_log.exists()
            assert "CHILD-LINE-MARKER" in child_log.read_text()
            master_log = tmp_path / "fam.log.MainProcess"
            if master_log.exists():
                assert "CHILD-LINE-MARKER" not in master_log.read_text()
        finally:
            fam_config.init()
            util.init_logger(fam_config.get_object(), "MainProcess")
