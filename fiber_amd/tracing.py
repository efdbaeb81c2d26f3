"""Lightweight tracing: roctx ranges visible in rocprofv3 timelines.

The reference has no tracing at all; here the hot phases
(pool dispatch, ES rollout/gradient/collective) emit roctx ranges so
``rocprofv3 --marker-trace`` attributes GPU time to framework phases.
No-ops cleanly when libroctx is unavailable (CPU-only boxes).

Usage:
    from fiber_amd import tracing
    with tracing.range("es.rollout"):
        ...
"""

import contextlib
import ctypes
import os

_lib = None
_checked = False


def _load():
    global _lib, _checked
    if _checked:
        return _lib
    _checked = True
    if os.environ.get("FAM_DISABLE_ROCTX"):
        return None
    for name in ("libroctx64.so", "libroctx64.so.4",
                 "/opt/rocm/lib/libroctx64.so"):
        try:
            lib = ctypes.CDLL(name)
            lib.roctxRangePushA.argtypes = [ctypes.c_char_p]
            lib.roctxRangePop.argtypes = []
            _lib = lib
            break
        except OSError:
            continue
    return _lib


def push(name):
    lib = _load()
    if lib is not None:
        lib.roctxRangePushA(name.encode())


def pop():
    lib = _load()
    if lib is not None:
        lib.roctxRangePop()


@contextlib.contextmanager
def range(name):  # noqa: A001 - mirrors roctx naming
    push(name)
    try:
        yield
    finally:
        pop()


def available():
    return _load() is not None
