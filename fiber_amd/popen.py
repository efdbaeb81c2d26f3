"""Master side of the spawn protocol.

Equivalent role to reference ``fiber/popen_fiber_spawn.py`` (one background
accept-thread per master multiplexing all children by ident; three pickled
blobs shipped to each child), re-based on Unix-domain sockets — the
single-node admin channel (SURVEY §2c "TCP + random-port bind" row: UDS, no
port management).

Wire protocol (all messages are 4-byte big-endian length + payload):
  child → master:  ident (utf-8)
  master → child:  prep_data pickle, process_obj pickle, post_data pickle
The admin socket then stays open: it is the child-liveness sentinel on the
master side and the orphan-watchdog fd on the child side.
"""

import os
import socket
import struct
import sys
import threading
import time

from . import backend as fam_backend
from . import config as fam_config
from . import serialization, util
from .core import JobSpec

_FAM_WORKER_ENV = "FAM_WORKER"


def send_msg(sock, payload):
    sock.sendall(struct.pack(">I", len(payload)) + payload)


def recv_msg(sock):
    header = _recv_exact(sock, 4)
    if header is None:
        return None
    (length,) = struct.unpack(">I", header)
    return _recv_exact(sock, length)


def _recv_exact(sock, n):
    chunks = []
    got = 0
    while got < n:
        chunk = sock.recv(n - got)
        if not chunk:
            return None
        chunks.append(chunk)
        got += len(chunk)
    return b"".join(chunks)


class _AdminListener:
    """One per master process: accepts child dial-backs, routes by ident."""

    def __init__(self):
        conf = fam_config.get_object()
        self.path = util.ipc_path(
            conf, util.random_name("fam-admin-%d" % os.getpid())
        )
        self._sock = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
        self._sock.bind(self.path)
        self._sock.listen(512)
        self._events = {}
        self._conns = {}
        self._lock = threading.Lock()
        self._thread = threading.Thread(
            target=self._accept_loop, name="fam_admin", daemon=True
        )
        self._thread.start()

    def expect(self, ident):
        event = threading.Event()
        with self._lock:
            self._events[ident] = event
        return event

    def take(self, ident):
        with self._lock:
            self._events.pop(ident, None)
            return self._conns.pop(ident, None)

    def _accept_loop(self):
        while True:
            try:
                conn, _ = self._sock.accept()
            except OSError:
                return
            threading.Thread(
                target=self._handshake, args=(conn,), daemon=True
            ).start()

    def _handshake(self, conn):
        try:
            ident_bytes = recv_msg(conn)
            if ident_bytes is None:
                conn.close()
                return
            ident = ident_bytes.decode()
            with self._lock:
                event = self._events.get(ident)
                if event is None:
                    conn.close()
                    return
                self._conns[ident] = conn
            event.set()
        except OSError:
            conn.close()


_listener = None
_listener_lock = threading.Lock()


def get_admin_listener():
    global _listener
    with _listener_lock:
        if _listener is None or not os.path.exists(_listener.path):
            _listener = _AdminListener()
        return _listener


def _build_prep_data():
    """multiprocessing-compatible preparation data (minus authkey)."""
    from multiprocessing import spawn as mp_spawn

    try:
        data = mp_spawn.get_preparation_data("fam_child")
    except RuntimeError:
        data = {
            "sys_path": sys.path,
            "sys_argv": sys.argv,
            "cwd": os.getcwd(),
        }
    data.pop("authkey", None)
    # The child applies our config, not a stdlib start-method.
    data.pop("start_method", None)
    # Drop un-importable synthetic main paths ('<stdin>', deleted files).
    main_path = data.get("init_main_from_path")
    if main_path is not None and (
        os.path.basename(main_path).startswith("<")
        or not os.path.exists(main_path)
    ):
        data.pop("init_main_from_path", None)
    data["fam_config"] = fam_config.get_dict()
    return data


class Popen:
    def __init__(self, process_obj):
        self._job = None
        self._conn = None
        self._exitcode = None
        self._backend = fam_backend.get_backend()
        self._launch(process_obj)

    def _launch(self, process_obj):
        conf = fam_config.get_object()
        listener = get_admin_listener()
        ident = util.random_name("p")
        event = listener.expect(ident)

        command = [
            sys.executable,
            "-m",
            "fiber_amd.spawn",
            listener.path,
            ident,
        ]
        meta = getattr(process_obj._target, "__fiber_meta__", None) or {}
        spec = JobSpec(
            command=command,
            name=process_obj.name,
            cpu=meta.get("cpu", conf.cpu_per_job),
            gpu=meta.get("gpu", conf.gpu_per_job),
            mem=meta.get("memory"),
            env={_FAM_WORKER_ENV: "1"},
        )
        self._job = self._backend.create_job(spec)

        deadline = time.monotonic() + conf.start_timeout
        while not event.wait(timeout=0.1):
            if time.monotonic() > deadline:
                listener.take(ident)
                self._backend.terminate_job(self._job)
                raise TimeoutError(
                    "worker %s did not dial back within %.0fs"
                    % (process_obj.name, conf.start_timeout)
                )
            # Fail fast if the job already died (e.g. bad interpreter).
            code = self._backend.get_job_exitcode(self._job)
            if code is not None:
                listener.take(ident)
                logs = ""
                try:
                    logs = self._backend.get_job_logs(self._job)[-2000:]
                except Exception:
                    pass
                raise RuntimeError(
                    "worker %s exited with code %s before handshake; "
                    "logs:\n%s" % (process_obj.name, code, logs)
                )
        self._conn = listener.take(ident)

        interactive = util.is_in_interactive_console()
        send_msg(self._conn, serialization.dumps(_build_prep_data()))
        if interactive:
            payload = serialization.dumps(process_obj, interactive=True)
        else:
            payload = serialization.dumps_closure(process_obj)
        send_msg(self._conn, payload)
        send_msg(self._conn, serialization.dumps({"pid": self._job.data.pid}))

    @property
    def pid(self):
        return self._job.data.pid

    @property
    def sentinel(self):
        return self._conn.fileno()

    def poll(self):
        if self._exitcode is None:
            self._exitcode = self._backend.get_job_exitcode(self._job)
            if self._exitcode is not None:
                self._reap_log()
        return self._exitcode

    def _reap_log(self):
        """Remove the job's capture file once it exited cleanly & silent."""
        path = getattr(self._job, "log_path", None)
        if path and self._exitcode == 0:
            try:
                if os.path.getsize(path) == 0:
                    os.unlink(path)
                    self._job.log_path = None
            except OSError:
                pass

    def wait(self, timeout=None):
        if self._exitcode is not None:
            return self._exitcode
        self._exitcode = self._backend.wait_for_job(self._job, timeout)
        if self._exitcode is not None:
            self._reap_log()
        return self._exitcode

    def get_logs(self):
        return self._backend.get_job_logs(self._job)

    def terminate(self):
        self._backend.terminate_job(self._job)

    def kill(self):
        kill = getattr(self._backend, "kill_job", None)
        if kill is not None:
            kill(self._job)
        else:
            self._backend.terminate_job(self._job)

    def close(self):
        if self._conn is not None:
            try:
                self._conn.close()
            except OSError:
                pass
            self._conn = None
