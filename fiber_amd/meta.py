"""``@fiber_amd.meta`` resource-hint decorator.

Parity with the reference decorator (uber/fiber ``fiber/meta.py:28-58``):
attaches a ``__fiber_meta__`` dict to the target function; the spawn layer
merges it into the JobSpec.  MI355X-native keys: ``gpu`` means "pin this
worker to a dedicated MI355X device" (the local backend assigns device
ordinals round-robin and exports ``HIP_VISIBLE_DEVICES``).
"""

import functools

VALID_META_KEYS = ("cpu", "memory", "gpu")


def meta(**kwargs):
    for key in kwargs:
        if key not in VALID_META_KEYS:
            raise ValueError(
                "invalid meta key %r; valid keys: %s" % (key, VALID_META_KEYS)
            )

    def decorator(func):
        @functools.wraps(func)
        def wrapper(*args, **kw):
            return func(*args, **kw)

        wrapper.__fiber_meta__ = dict(kwargs)
        return wrapper

    return decorator


def get_meta(func):
    """Return the resource-hint dict attached to *func* (or None)."""
    return getattr(func, "__fiber_meta__", None)
