"""Drop-in import path for reference users:
``from fiber.experimental.ring import Ring`` becomes
``from fiber_amd.experimental.ring import Ring``.

The Ring itself is no longer experimental here — it is the native
collective engine (see fiber_amd/ring.py).
"""

from ..ring import Ring, RingContext, RingNode  # noqa: F401
