"""Raw pool dispatch rate (noop tasks) at several chunksizes."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from fiber_amd.pool import ZPool


def noop(x):
    return x


def main():
    p = ZPool(processes=8)
    try:
        p.map(noop, range(1000), chunksize=32)  # warm workers
        for cs in (1, 8, 32):
            n = 40000 if cs == 1 else 200000
            t0 = time.perf_counter()
            r = p.map(noop, range(n), chunksize=cs)
            dt = time.perf_counter() - t0
            assert len(r) == n
            print("chunksize %2d: %9.0f tasks/s (%d tasks, %.2f s)"
                  % (cs, n / dt, n, dt))
    finally:
        p.terminate()
        p.join()


if __name__ == "__main__":
    main()
