"""Flagship ES running THROUGH the framework: Pool fan-out + collectives.

The reference's identity workload is "ES via pool.map"
(uber/fiber examples/gecco-2020/es.py:17-34: pool.map of perturbation
rollouts, centered-rank normalize).  This is the MI355X-native form:

* the Pool owns an RCCL communicator over its GPU-pinned workers
  (``collective=True``; one worker == one MI355X);
* each worker holds a persistent :class:`fiber_amd.es.ESEngine` (built in
  the pool initializer) whose ``ctx`` IS the pool's worker group — the
  per-iteration fitness all-gather and gradient all-reduce run
  worker-to-worker over xGMI, not through the master;
* the master fan-outs via :meth:`ZPool.run_on_all` — one SPMD call per
  worker per block of iterations — and fan-ins the scalar stats over the
  shm rings.  Iteration blocks amortize the ~ms dispatch latency to
  nothing (see profiles/r02_es_pool_overhead.md).

CPU demo (stub rollouts, gloo, 2 workers):
    python examples/es_pool.py --cpu-demo
GPU (one worker per visible MI355X):
    python examples/es_pool.py --iters 50 --block 10
"""

import os as _os
import sys as _sys

_REPO_ROOT = _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__)))
if _REPO_ROOT not in _sys.path:
    _sys.path.insert(0, _REPO_ROOT)

import argparse
import time

_ENGINE = None


def init_es_worker(cfg_kwargs, model="mlp", stub=False):
    """Pool initializer: build the persistent per-worker ES engine, with
    the pool's worker group as its collective context."""
    global _ENGINE
    import torch

    from fiber_amd.pool import current_worker_group

    group = current_worker_group()
    if stub:  # CPU plumbing demo/tests: replace HIP kernels
        from fiber_amd import ops
        from tests.test_es_distributed import _cpu_stub_ops

        _cpu_stub_ops(ops)
        device = torch.device("cpu")
    else:
        device = group.device
    ctx = group if group.size > 1 else None
    if model == "mlp":
        from fiber_amd.es import ESConfig, ESEngine

        _ENGINE = ESEngine(ESConfig(**cfg_kwargs), ctx=ctx, device=device)
    else:
        from fiber_amd.es.conv_policy import ConvESConfig, ConvESEngine

        _ENGINE = ConvESEngine(ConvESConfig(**cfg_kwargs), ctx=ctx,
                               device=device)


def es_steps(start_iter, k, sync=False):
    """Run k ES iterations on this worker's shard; returns the last
    iteration's stats dict (plus theta checksum for tests)."""
    stats = None
    for i in range(k):
        stats = _ENGINE.step(iteration=start_iter + i)
    if sync:
        import torch

        if _ENGINE.device.type == "cuda":
            torch.cuda.synchronize()
    stats = dict(stats)
    stats["theta_sum"] = float(_ENGINE.theta.double().sum())
    return stats


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--iters", type=int, default=20)
    parser.add_argument("--block", type=int, default=5,
                        help="iterations per SPMD dispatch")
    parser.add_argument("--pop-per-gpu", type=int, default=None)
    parser.add_argument("--horizon", type=int, default=None)
    parser.add_argument("--model", choices=["mlp", "conv"], default="mlp")
    parser.add_argument("--workers", type=int, default=None)
    parser.add_argument("--cpu-demo", action="store_true")
    args = parser.parse_args()

    import fiber_amd
    from fiber_amd.pool import ZPool

    if args.cpu_demo:
        workers = args.workers or 2
        cfg = dict(pop_per_gpu=args.pop_per_gpu or 8,
                   horizon=args.horizon or 4, seed=99)
        stub = True
        gpu_per_worker = None
    else:
        workers = args.workers or max(1, fiber_amd.gpu_count())
        cfg = dict(
            pop_per_gpu=args.pop_per_gpu
            or (32768 if args.model == "mlp" else 2048),
            horizon=args.horizon or (256 if args.model == "mlp" else 64),
        )
        stub = False
        gpu_per_worker = 1

    pool = ZPool(processes=workers, gpu_per_worker=gpu_per_worker,
                 collective=True,
                 initializer=init_es_worker,
                 initargs=(cfg, args.model, stub))
    try:
        done = 0
        t0 = time.perf_counter()
        while done < args.iters:
            k = min(args.block, args.iters - done)
            per_rank = pool.run_on_all(es_steps, (done, k, True))
            done += k
            s = per_rank[0]
            print("iter %4d  fitness %8.4f  (max %8.4f)  theta %.6f"
                  % (done, s["fitness_mean"], s["fitness_max"],
                     s["theta_sum"]))
            assert all(abs(r["theta_sum"] - s["theta_sum"]) < 1e-9
                       for r in per_rank), "rank theta divergence"
        wall = time.perf_counter() - t0
        total = per_rank[0]["rollouts"] * args.iters
        print("%.1f rollouts/s through the pool (%d iters, %d workers)"
              % (total / wall, args.iters, workers))
    finally:
        pool.terminate()
        pool.join()


if __name__ == "__main__":
    main()
