"""Queue throughput driver (reference examples/bench_queue.py analog):
N messages through SimpleQueue across two processes, msgs/sec + MB/s,
fiber_amd shm rings vs stdlib multiprocessing.
"""

import os as _os
import sys as _sys

_REPO_ROOT = _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__)))
if _REPO_ROOT not in _sys.path:
    _sys.path.insert(0, _REPO_ROOT)


import argparse
import multiprocessing
import time


def _fam_echo_sink(q, out, n):
    for _ in range(n):
        q.get()
    out.put("done")


def _mp_echo_sink(q, out, n):
    for _ in range(n):
        q.get()
    out.put("done")


def run_fiber_amd(n, payload):
    import fiber_amd
    from fiber_amd.queues import SimpleQueue

    q, out = SimpleQueue(), SimpleQueue()
    p = fiber_amd.Process(target=_fam_echo_sink, args=(q, out, n))
    p.start()
    t0 = time.perf_counter()
    for _ in range(n):
        q.put(payload)
    out.get(timeout=600)
    elapsed = time.perf_counter() - t0
    p.join(30)
    q.close()
    out.close()
    return elapsed


def run_multiprocessing(n, payload):
    q = multiprocessing.SimpleQueue()
    out = multiprocessing.SimpleQueue()
    p = multiprocessing.Process(target=_mp_echo_sink, args=(q, out, n))
    p.start()
    t0 = time.perf_counter()
    for _ in range(n):
        q.put(payload)
    out.get()
    elapsed = time.perf_counter() - t0
    p.join(30)
    return elapsed


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("-n", type=int, default=200_000)
    parser.add_argument("--size", type=int, default=100)
    args = parser.parse_args()
    payload = b"x" * args.size

    fam = run_fiber_amd(args.n, payload)
    mp_t = run_multiprocessing(args.n, payload)
    for name, elapsed in (("fiber_amd", fam), ("multiprocessing", mp_t)):
        rate = args.n / elapsed
        mbps = rate * args.size / 1e6
        print("%-16s %8.0f msgs/s  %7.1f MB/s  (%.2fs for %d x %dB)"
              % (name, rate, mbps, elapsed, args.n, args.size))
    print("speedup: %.2fx" % (mp_t / fam))


if __name__ == "__main__":
    main()
