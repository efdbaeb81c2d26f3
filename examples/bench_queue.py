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


def _fam_batch_sink(q, out, n, batch):
    got = 0
    while got < n:
        got += len(q.get_many(max_n=batch, timeout=60.0))
    out.put("done")


def run_fiber_amd_batched(n, payload, batch=128):
    """put_many/get_many path: one lock hold + one wake per burst."""
    import fiber_amd
    from fiber_amd.queues import SimpleQueue

    q, out = SimpleQueue(), SimpleQueue()
    p = fiber_amd.Process(target=_fam_batch_sink, args=(q, out, n, batch))
    p.start()
    burst = [payload] * batch
    t0 = time.perf_counter()
    left = n
    while left > 0:
        take = burst if left >= batch else burst[:left]
        q.put_many(take)
        left -= len(take)
    out.get(timeout=600)
    elapsed = time.perf_counter() - t0
    p.join(30)
    q.close()
    out.close()
    return elapsed


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
    parser.add_argument("--batch", type=int, default=0,
                        help="also run the put_many/get_many path with "
                             "this burst size (e.g. 128)")
    args = parser.parse_args()
    payload = b"x" * args.size

    rows = [("fiber_amd", run_fiber_amd(args.n, payload))]
    if args.batch:
        rows.append(("fiber_amd batch=%d" % args.batch,
                     run_fiber_amd_batched(args.n, payload, args.batch)))
    mp_t = run_multiprocessing(args.n, payload)
    rows.append(("multiprocessing", mp_t))
    for name, elapsed in rows:
        rate = args.n / elapsed
        mbps = rate * args.size / 1e6
        print("%-20s %8.0f msgs/s  %7.1f MB/s  (%.2fs for %d x %dB)"
              % (name, rate, mbps, elapsed, args.n, args.size))
    print("speedup: %.2fx" % (mp_t / rows[0][1]))


if __name__ == "__main__":
    main()
