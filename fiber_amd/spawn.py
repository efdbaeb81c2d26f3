"""Worker-side spawn entry: ``python -m fiber_amd.spawn <sock> <ident>``.

Equivalent role to reference ``fiber/spawn.py:33-82`` (spawn_prepare +
orphan watchdog): connect back to the master's admin Unix socket, receive
prep data (config + interpreter state), the pickled Process, and post data,
then run ``Process._bootstrap()``.  A daemon watchdog thread selects on the
admin socket and SIGTERMs this process when the master vanishes.
"""

import os
import select
import signal
import socket
import sys
import threading


def _watchdog(sock):
    """Exit when the admin socket hits EOF (master died or closed us)."""
    while True:
        try:
            readable, _, _ = select.select([sock], [], [], 1.0)
        except (OSError, ValueError):
            break
        if readable:
            try:
                data = sock.recv(1, socket.MSG_PEEK)
            except OSError:
                break
            if not data:
                break
    os.kill(os.getpid(), signal.SIGTERM)


def spawn_main(sock_path, ident):
    from . import config as fam_config
    from .popen import recv_msg, send_msg
    from . import serialization

    sock = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
    sock.connect(sock_path)
    send_msg(sock, ident.encode())

    prep_data = serialization.loads(recv_msg(sock))
    fam_conf = prep_data.pop("fam_config", {})
    fam_config.init(**{k: v for k, v in fam_conf.items() if v is not None})

    from multiprocessing import spawn as mp_spawn

    prep_data.pop("authkey", None)
    prep_data.pop("start_method", None)
    try:
        mp_spawn.prepare(prep_data)
    except Exception:
        # Main-module re-import is best-effort: closures shipped via
        # cloudpickle do not need it.
        import traceback

        traceback.print_exc()

    process_obj = serialization.loads(recv_msg(sock))
    serialization.loads(recv_msg(sock))  # post_data (reserved)

    thread = threading.Thread(target=_watchdog, args=(sock,), daemon=True)
    thread.start()

    process_obj._pid = os.getpid()
    exitcode = process_obj._bootstrap()
    sys.exit(exitcode)


if __name__ == "__main__":
    spawn_main(sys.argv[1], sys.argv[2])
