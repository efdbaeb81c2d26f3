"""ES-through-Pool (the flagship through the framework).

Equivalence oracle: the same ES evolution run (a) through the Pool SPMD
path (run_on_all + pool-communicator collectives) and (b) through the
direct Ring path (tests/test_es_distributed.py) from identical seeds
must produce bit-identical theta — the framework adds dispatch, not
numerics.
"""

import functools

import torch

from fiber_amd.pool import ZPool
from fiber_amd.queues import SimpleQueue
from fiber_amd.ring import Ring, RingContext
from tests.test_es_distributed import _cpu_stub_ops

from examples.es_pool import es_steps, init_es_worker

_CFG = dict(pop_per_gpu=8, horizon=4, seed=99)


def _ring_rank_main(rank, size, out=None):
    from fiber_amd import ops
    from fiber_amd.es import ESConfig, ESEngine

    _cpu_stub_ops(ops)
    ctx = RingContext(rank, size, backend="gloo",
                      device=torch.device("cpu"))
    ctx.init()
    engine = ESEngine(ESConfig(**_CFG), ctx=ctx, device=torch.device("cpu"))
    for i in range(3):
        engine.step(iteration=i)
    out.put((rank, float(engine.theta.double().sum())))
    ctx.shutdown()


class TestESThroughPool:
    def test_pool_path_matches_ring_path(self):
        # (a) through the Pool
        pool = ZPool(processes=2, collective=True,
                     initializer=init_es_worker,
                     initargs=(_CFG, "mlp", True))
        try:
            per_rank = pool.run_on_all(es_steps, (0, 3), timeout=300)
        finally:
            pool.terminate()
            pool.join()
        pool_thetas = [r["theta_sum"] for r in per_rank]
        assert pool_thetas[0] == pool_thetas[1]  # ranks identical

        # (b) through the Ring (the direct path the bench certifies)
        out = SimpleQueue()
        ring = Ring(2, functools.partial(_ring_rank_main, out=out),
                    backend="gloo", gpu_per_rank=0)
        ring.run(timeout=300)
        ring_thetas = [out.get(timeout=10)[1] for _ in range(2)]
        out.close()
        assert ring_thetas[0] == ring_thetas[1]

        # the framework adds dispatch, not numerics
        assert pool_thetas[0] == ring_thetas[0]

    def test_stats_fan_in(self):
        pool = ZPool(processes=2, collective=True,
                     initializer=init_es_worker,
                     initargs=(_CFG, "mlp", True))
        try:
            a = pool.run_on_all(es_steps, (0, 1), timeout=300)
            b = pool.run_on_all(es_steps, (1, 2), timeout=300)
        finally:
            pool.terminate()
            pool.join()
        # every rank reports the same full-population fitness
        assert a[0]["fitness_mean"] == a[1]["fitness_mean"]
        assert b[0]["fitness_mean"] == b[1]["fitness_mean"]
        # rollout accounting covers the whole population
        assert a[0]["rollouts"] == _CFG["pop_per_gpu"] * 2 * 64
