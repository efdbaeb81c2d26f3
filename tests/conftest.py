import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU (run with -m gpu)"
    )


@pytest.fixture(autouse=True)
def leak_check():
    """Every test doubles as a process/shm leak detector (reference idiom:
    uber/fiber tests/test_pool.py:75-84 autouse fixture)."""
    import fiber_amd

    assert fiber_amd.active_children() == []
    before = set(os.listdir("/dev/shm")) if os.path.isdir("/dev/shm") else set()
    yield
    # Children must be reaped by each test (allow a short grace period
    # for SIGTERM delivery on just-killed workers).
    import time

    deadline = time.monotonic() + 10.0
    leftover = fiber_amd.active_children()
    while leftover and time.monotonic() < deadline:
        time.sleep(0.05)
        leftover = fiber_amd.active_children()
    for proc in leftover:
        proc.terminate()
        proc.join(5)
    assert leftover == [], "leaked child processes: %r" % leftover
    import gc

    gc.collect()
    after = set(os.listdir("/dev/shm")) if os.path.isdir("/dev/shm") else set()
    new = {n for n in (after - before) if n.startswith("fam-")}
    if new:
        # Under load, a just-reaped child's normal teardown (atexit
        # close+unlink) can still be in flight when the test returns.
        # Give it a short settle window; anything that PERSISTS has no
        # owner left to unlink it and is a real leak.
        settle = time.monotonic() + 3.0
        while new and time.monotonic() < settle:
            time.sleep(0.1)
            still = set(os.listdir("/dev/shm"))
            new = {n for n in new if n in still}
    for name in new:
        try:
            os.unlink(os.path.join("/dev/shm", name))
        except OSError:
            pass
    # Per-worker reply rings are worker-owned and SIGKILL-racy by design
    # (a core forked during shutdown can create one after the master's
    # sweep); they are cleaned above but not a test failure.
    new = {n for n in new if ".task.r." not in n}
    assert not new, "leaked shm segments: %r" % new


# On-demand all-thread stack dumps for hang forensics:
#   FAM_DEBUG_FAULTHANDLER=1 pytest ... ; kill -USR1 <pid>
if os.environ.get("FAM_DEBUG_FAULTHANDLER"):
    import faulthandler
    import signal as _signal

    faulthandler.register(_signal.SIGUSR1, all_threads=True)
