"""Monte-Carlo pi estimation on the Pool — BASELINE config 1.

Reference analog: uber/fiber examples/pi_estimation.py (Pool(4).map over
1e6 samples on the local backend, CPU plumbing only).
"""

import os as _os
import sys as _sys

_REPO_ROOT = _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__)))
if _REPO_ROOT not in _sys.path:
    _sys.path.insert(0, _REPO_ROOT)


import random
import time

import fiber_amd


def is_inside(seed):
    random.seed(seed)
    x, y = random.random(), random.random()
    return x * x + y * y < 1


def main(samples=1_000_000, processes=4):
    pool = fiber_amd.Pool(processes)
    try:
        t0 = time.perf_counter()
        hits = sum(pool.map(is_inside, range(samples), chunksize=2048))
        elapsed = time.perf_counter() - t0
        pi = 4.0 * hits / samples
        print("pi ~= %.5f  (%d samples, %d workers, %.2fs, %.0f tasks/s)"
              % (pi, samples, processes, elapsed, samples / elapsed))
        return pi
    finally:
        pool.terminate()
        pool.join()


if __name__ == "__main__":
    main()
