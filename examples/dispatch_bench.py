"""Pool.map task-dispatch overhead microbench — BASELINE config 2.

Reference analog: uber/fiber examples/bench_frameworks.py — a batch of
tasks whose ideal wall time is 1 s on N workers, swept over task
durations 1 s -> 1 ms.  Fiber's headline claim is "~= multiprocessing
overhead at >=100 ms tasks, graceful at 1 ms"; this driver reports the
same overhead ratio for fiber_amd's shm-ring pool vs. stdlib
multiprocessing.Pool on the same machine.

Run: python examples/dispatch_bench.py [--workers 8]
"""

import os as _os
import sys as _sys

_REPO_ROOT = _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__)))
if _REPO_ROOT not in _sys.path:
    _sys.path.insert(0, _REPO_ROOT)


import argparse
import multiprocessing
import time


def sleep_task(duration):
    time.sleep(duration)
    return duration


def _drive(pool, duration, n_tasks, chunksize):
    t0 = time.perf_counter()
    pool.map(sleep_task, [duration] * n_tasks, chunksize=chunksize)
    return time.perf_counter() - t0


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--workers", type=int, default=8)
    parser.add_argument("--ideal-seconds", type=float, default=1.0)
    args = parser.parse_args()
    durations = [1.0, 0.1, 0.01, 0.001]

    import fiber_amd

    rows = []
    for duration in durations:
        n_tasks = max(args.workers,
                      int(args.ideal_seconds * args.workers / duration))
        chunksize = max(1, n_tasks // (args.workers * 8))

        fam_pool = fiber_amd.Pool(args.workers)
        try:
            # Warm until every worker has booted (interpreter start is
            # ~150 ms) — parallel sleep tasks force full participation.
            fam_pool.map(sleep_task, [0.2] * args.workers, chunksize=1)
            fam_wall = _drive(fam_pool, duration, n_tasks, chunksize)
        finally:
            fam_pool.terminate()
            fam_pool.join()

        mp_pool = multiprocessing.Pool(args.workers)
        try:
            mp_pool.map(sleep_task, [0.2] * args.workers, chunksize=1)
            mp_wall = _drive(mp_pool, duration, n_tasks, chunksize)
        finally:
            mp_pool.terminate()
            mp_pool.join()

        ideal = n_tasks * duration / args.workers
        rows.append((duration, n_tasks, ideal, fam_wall, mp_wall))
        print("task=%7.3fs n=%6d ideal=%5.2fs  fiber_amd=%6.3fs (%.2fx)  "
              "multiprocessing=%6.3fs (%.2fx)"
              % (duration, n_tasks, ideal, fam_wall, fam_wall / ideal,
                 mp_wall, mp_wall / ideal))
    return rows


if __name__ == "__main__":
    main()
