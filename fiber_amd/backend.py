"""Backend registry: name → singleton instance.

Parity with reference ``fiber/backend.py:24-76``.  Only ``local`` ships;
the registry keeps the seam so tests can hot-swap fault-injecting backends
(the reference test idiom, SURVEY §4 "fault injection by subclassing the
Backend").
"""

import importlib
import threading

from . import config as fam_config

available_backend = ["local"]

_backends = {}
_lock = threading.Lock()


def auto_select_backend():
    conf = fam_config.get_object()
    if conf.backend:
        return conf.backend
    return conf.default_backend or "local"


def get_backend(name=None):
    if name is None:
        name = auto_select_backend()
    with _lock:
        if name in _backends:
            return _backends[name]
        if name not in available_backend:
            raise ValueError(
                "unknown backend %r; available: %s" % (name, available_backend)
            )
        module = importlib.import_module("fiber_amd.backends." + name)
        inst = module.Backend()
        _backends[name] = inst
        return inst


def set_backend(name, instance):
    """Install a backend instance under *name* (test fault-injection seam)."""
    with _lock:
        _backends[name] = instance
        if name not in available_backend:
            available_backend.append(name)


def reset():
    with _lock:
        _backends.clear()
