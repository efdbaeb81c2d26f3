"""fiber_amd — an MI355X-native distributed worker-pool framework.

A from-scratch rebuild of the capability surface of uber/fiber
(multiprocessing-compatible Process / Pool / SimpleQueue / Pipe / Manager /
Ring / meta / config — see SURVEY.md §2), designed for a single node of
AMD Instinct MI355X GPUs:

* the data plane is a C++ shared-memory ring engine (``fiber_amd.transport``)
  instead of nanomsg TCP;
* tensor-typed task args/results stay device-resident and cross process
  boundaries as HIP IPC handles (``fiber_amd.serialization``);
* ``Ring`` owns a real collective engine (RCCL over xGMI via
  ``torch.distributed``) rather than being rendezvous-only;
* the ES hot path (batched policy rollouts, observation normalisation,
  centered-rank, noise-table gradients) is hand-written CDNA4 HIP
  (``fiber_amd.ops``).

``import fiber_amd as mp`` is a drop-in for ``multiprocessing`` to the same
degree the reference is ("one-line fiberization").
"""

import os

__version__ = "0.2.0"

from . import config as _config_mod
from .config import Config  # noqa: F401
from .meta import meta  # noqa: F401
from .process import (  # noqa: F401
    Process,
    active_children,
    current_process,
)


def init(**kwargs):
    """(Re-)initialize config and logging from file/env/kwargs."""
    conf = _config_mod.init(**kwargs)
    from . import util

    util.init_logger(conf, current_process().name)
    return conf


def reset():
    """Reset config to defaults and clear backend singletons."""
    from . import backend as _backend_mod

    conf = _config_mod.init()
    _backend_mod.reset()
    return conf


def in_worker():
    return os.environ.get("FAM_WORKER") == "1"


# ---------------------------------------------------------------------------
# Context façade (reference parity: fiber/context.py — factory methods for
# the coordination primitives; only the spawn start method exists).
# ---------------------------------------------------------------------------


def cpu_count():
    return os.cpu_count()


def gpu_count():
    try:
        import torch

        return torch.cuda.device_count()
    except Exception:
        return 0


def Pool(processes=None, initializer=None, initargs=(), maxtasksperchild=None,
         error_handling=False, **kwargs):
    """Pool factory (reference ``fiber/context.py:38-45`` semantics).

    ``error_handling=False`` (default, same as the reference context
    factory) returns :class:`~fiber_amd.pool.ZPool`: worker exceptions
    are delivered to the caller.  ``error_handling=True`` returns
    :class:`~fiber_amd.pool.ResilientZPool` (also exposed as
    ``fiber_amd.pool.Pool``, matching the reference's module alias):
    a failing worker is killed and its pending chunks are resubmitted —
    tasks must be idempotent.  The two entry points deliberately differ,
    exactly as in the reference.
    """
    from . import pool as _pool_mod

    if error_handling:
        cls = _pool_mod.ResilientZPool
    else:
        cls = _pool_mod.ZPool
    return cls(
        processes=processes,
        initializer=initializer,
        initargs=initargs,
        maxtasksperchild=maxtasksperchild,
        **kwargs,
    )


def SimpleQueue():
    from . import queues as _queues_mod

    return _queues_mod.SimpleQueue()


def Pipe(duplex=True):
    from . import queues as _queues_mod

    return _queues_mod.Pipe(duplex=duplex)


def Manager():
    from . import managers as _managers_mod

    manager = _managers_mod.SyncManager()
    manager.start()
    return manager


def AsyncManager():
    from . import managers as _managers_mod

    manager = _managers_mod.AsyncManager()
    manager.start()
    return manager


def Ring(size, func, initializer=None, gpu_per_rank=None, backend=None):
    from . import ring as _ring_mod

    return _ring_mod.Ring(
        size, func, initializer=initializer, gpu_per_rank=gpu_per_rank,
        backend=backend,
    )


def get_context(method=None):
    """Only the 'spawn' start method exists (reference parity)."""
    if method not in (None, "spawn"):
        raise ValueError("fiber_amd only supports the 'spawn' start method")
    import fiber_amd

    return fiber_amd
