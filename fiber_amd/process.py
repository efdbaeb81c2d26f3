"""Job-backed Process with the ``multiprocessing.Process`` API.

Parity target: reference ``fiber/process.py:83-323`` (start/join/is_alive/
terminate/sentinel/active_children/current_process semantics), rebuilt
without subclassing stdlib internals: a ``Process`` here is a plain object
whose ``start()`` launches a backend job (a pinned subprocess on the
single-node backend) and ships the pickled Process to it over a Unix-socket
admin channel (see ``fiber_amd.popen``).
"""

import itertools
import os
import threading

from . import util

_children = set()
_children_lock = threading.Lock()
_process_counter = itertools.count(1)


class _CurrentProcess:
    """Lightweight stand-in describing the running process."""

    def __init__(self, name="MainProcess"):
        self.name = name
        self.daemon = False

    @property
    def pid(self):
        return os.getpid()

    @property
    def exitcode(self):
        return None

    def is_alive(self):
        return True

    def __repr__(self):
        return "<%s name=%r pid=%d>" % (
            type(self).__name__,
            self.name,
            os.getpid(),
        )


_current = _CurrentProcess()


def current_process():
    return _current


def _set_current_process(proc):
    global _current
    _current = proc


def _cleanup():
    with _children_lock:
        dead = [p for p in _children if p._popen and p._popen.poll() is not None]
        for p in dead:
            _children.discard(p)


def active_children():
    """Live child Process objects (reaps finished ones first)."""
    _cleanup()
    with _children_lock:
        return [p for p in _children if p.is_alive()]


class Process:
    def __init__(
        self,
        group=None,
        target=None,
        name=None,
        args=(),
        kwargs=None,
        daemon=None,
    ):
        if group is not None:
            raise ValueError("group argument must be None")
        self._target = target
        self._args = tuple(args)
        self._kwargs = dict(kwargs or {})
        count = next(_process_counter)
        self._name = name or ("Process-%d" % count)
        self.daemon = bool(daemon)
        self._popen = None
        self._pid = None
        self._closed = False

    # -- identity ----------------------------------------------------------
    @property
    def name(self):
        return self._name

    @name.setter
    def name(self, value):
        self._name = value

    @property
    def pid(self):
        return self._pid

    ident = pid

    @property
    def exitcode(self):
        if self._popen is None:
            return None
        return self._popen.poll()

    @property
    def sentinel(self):
        """A selectable fd that becomes readable when the child exits."""
        if self._popen is None:
            raise ValueError("process not started")
        return self._popen.sentinel

    # -- lifecycle ---------------------------------------------------------
    def start(self):
        if self._popen is not None:
            raise RuntimeError("cannot start a process twice")
        if self._closed:
            raise ValueError("process object is closed")
        _cleanup()
        from .popen import Popen

        self._popen = Popen(self)
        self._pid = self._popen.pid
        with _children_lock:
            _children.add(self)

    def join(self, timeout=None):
        if self._popen is None:
            raise RuntimeError("can only join a started process")
        code = self._popen.wait(timeout)
        if code is not None:
            with _children_lock:
                _children.discard(self)
        return None

    def is_alive(self):
        if self._popen is None:
            return False
        return self._popen.poll() is None

    def terminate(self):
        if self._popen is not None:
            self._popen.terminate()

    def kill(self):
        if self._popen is not None:
            self._popen.kill()

    def logs(self):
        """Captured stdout/stderr of the job (reference get_job_logs)."""
        if self._popen is None:
            return ""
        return self._popen.get_logs()

    def close(self):
        if self._popen is not None and self._popen.poll() is None:
            raise ValueError("cannot close a process while it is still running")
        if self._popen is not None:
            self._popen.close()
        self._closed = True

    # -- worker side -------------------------------------------------------
    def run(self):
        if self._target:
            self._target(*self._args, **self._kwargs)

    def _bootstrap(self):
        """Worker-side run loop; returns the exit code."""
        from . import config as fam_config

        _set_current_process(self)
        util.init_logger(fam_config.get_object(), self._name)
        log = util.get_logger()
        try:
            self.run()
            return 0
        except SystemExit as exc:
            code = exc.code
            if code is None:
                return 0
            if isinstance(code, int):
                return code
            return 1
        except Exception:
            import traceback

            traceback.print_exc()
            log.exception("process %s crashed", self._name)
            return 1

    # worker-side shims so a shipped Process behaves like current_process()
    def is_alive_worker(self):
        return True

    # -- pickling ----------------------------------------------------------
    def __reduce__(self):
        state = self.__dict__.copy()
        state["_popen"] = None
        return (_rebuild_process, (state,))

    def __repr__(self):
        status = "initial"
        if self._popen is not None:
            code = self._popen.poll()
            status = "started" if code is None else ("stopped[%s]" % code)
        return "<Process name=%r pid=%r %s>" % (self._name, self._pid, status)


def _rebuild_process(state):
    proc = Process.__new__(Process)
    proc.__dict__.update(state)
    return proc
