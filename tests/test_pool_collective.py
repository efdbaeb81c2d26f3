"""Pool RCCL data-plane tests (CPU plumbing: gloo, world 2).

The same code path carries RCCL over xGMI on GPU workers (backend
auto-selects "nccl"); these tests pin the choreography: one-shot shared
staging + broadcast, all-reduce fan-in, SPMD exec, and the
communicator-rebuild-on-death policy.
"""

import time

import pytest
import torch

from fiber_amd.pool import ZPool, ResilientZPool, current_worker_group


def _weighted(x, theta=None):
    # theta arrives via the pool broadcast, not via the chunk payload
    return float(theta.sum()) * x


def _onehot4(x):
    return torch.full((4,), float(x))


def _rank_world():
    g = current_worker_group()
    return (g.rank, g.size)


def _group_allreduce_rank():
    g = current_worker_group()
    t = torch.tensor([float(g.rank + 1)])
    g.allreduce(t)
    return float(t[0])


def _slow_echo(x):
    time.sleep(0.05)
    return x


@pytest.fixture
def cpool():
    p = ZPool(processes=2, collective=True)
    yield p
    p.terminate()
    p.join()


class TestSharedBroadcast:
    def test_shared_tensor_reaches_every_worker(self, cpool):
        theta = torch.arange(16, dtype=torch.float32)
        out = cpool.map(_weighted, range(8), chunksize=1,
                        shared={"theta": theta})
        want = [float(theta.sum()) * x for x in range(8)]
        assert out == want

    def test_shared_requires_collective_pool(self):
        with ZPool(processes=2) as p:
            with pytest.raises(ValueError):
                p.map(_weighted, range(4), shared={"theta": torch.ones(2)})

    def test_resilient_pool_rejects_collective(self):
        with pytest.raises(ValueError):
            ResilientZPool(processes=2, collective=True)


class TestReduceSum:
    def test_reduce_matches_plain_map(self, cpool):
        xs = list(range(10))
        reduced = cpool.map(_onehot4, xs, chunksize=2, reduce="sum",
                            reduce_spec=((4,), torch.float32))
        plain = cpool.map(_onehot4, xs, chunksize=2)
        want = torch.stack(plain).sum(dim=0)
        assert torch.equal(reduced, want)
        assert torch.equal(reduced, torch.full((4,), float(sum(xs))))

    def test_reduce_needs_spec(self, cpool):
        with pytest.raises(ValueError):
            cpool.map(_onehot4, range(4), reduce="sum")


class TestRunOnAll:
    def test_one_task_per_rank_ordered(self, cpool):
        out = cpool.run_on_all(_rank_world)
        assert out == [(0, 2), (1, 2)]

    def test_worker_group_collectives(self, cpool):
        # (rank0+1) + (rank1+1) = 3 on every rank
        out = cpool.run_on_all(_group_allreduce_rank)
        assert out == [3.0, 3.0]


class TestRebuildOnDeath:
    def test_collective_map_fails_fast_then_recovers(self, monkeypatch):
        # bound the rendezvous timeout so the rotation converges quickly
        # (workers inherit the master's env at spawn)
        monkeypatch.setenv("FAM_PG_TIMEOUT", "5")
        pool = ZPool(processes=2, collective=True)
        try:
            theta = torch.ones(4)
            res = pool.map_async(_slow_echo, range(60), chunksize=1,
                                 shared={"theta": theta})
            # let the map get going, then murder one worker
            time.sleep(0.6)
            with pool._worker_lock:
                victim = next(iter(pool._workers.values()))
            victim.kill()
            with pytest.raises(RuntimeError, match="collective"):
                res.get(60)
            # the group rebuilt (new generation): SPMD + collectives
            # work again.  Convergence needs a few attempts: each retry
            # that overlaps a rank still timing out on the previous
            # generation fails and rotates once more (bounded by the PG
            # timeout per attempt).
            deadline = time.monotonic() + 90
            while True:
                try:
                    out = pool.run_on_all(_group_allreduce_rank,
                                          timeout=60)
                    break
                except Exception:
                    if time.monotonic() > deadline:
                        raise
                    time.sleep(0.5)
            assert out == [3.0, 3.0]
        finally:
            pool.terminate()
            pool.join()


def _staged_maps_count():
    import fiber_amd.pool as pm

    return len(pm._current_coll_state.maps)


class TestStagedCleanup:
    def test_shared_tensors_freed_after_map(self, cpool):
        """Completed collective maps must drop their staged shared
        tensors in every worker (device memory on GPU pools) — the
        master broadcasts a drop on completion."""
        theta = torch.ones(64)
        for _ in range(3):
            cpool.map(_weighted, range(4), chunksize=1,
                      shared={"theta": theta})
        time.sleep(0.5)  # drop records are async to map completion
        counts = cpool.run_on_all(_staged_maps_count)
        assert counts == [0, 0], counts


class TestReduceEdges:
    def test_empty_iterable_reduce(self, cpool):
        out = cpool.map(_onehot4, [], reduce="sum",
                        reduce_spec=((4,), torch.float32))
        assert torch.equal(out, torch.zeros(4))

    def test_reduce_rejects_callbacks(self, cpool):
        with pytest.raises(ValueError, match="callbacks"):
            cpool.map_async(_onehot4, range(4), reduce="sum",
                            reduce_spec=((4,), torch.float32),
                            callback=lambda r: None)


def _slow_weighted(x, theta=None):
    time.sleep(0.01)
    return float(theta.sum()) * x


class TestConcurrentCollectives:
    def test_interleaved_shared_maps_and_spmd(self, cpool):
        """Multiple in-flight collective maps + SPMD execs: the ctl
        ordering guarantee must keep every rank's collective sequence
        identical (a mismatch deadlocks or mixes stage broadcasts)."""
        thetas = [torch.full((8,), float(k + 1)) for k in range(3)]
        asyncs = [
            cpool.map_async(_slow_weighted, range(6), chunksize=1,
                            shared={"theta": thetas[k]})
            for k in range(3)
        ]
        spmd = cpool.run_on_all(_group_allreduce_rank, timeout=120)
        assert spmd == [3.0, 3.0]
        for k, res in enumerate(asyncs):
            want = [float(thetas[k].sum()) * x for x in range(6)]
            assert res.get(120) == want
        red = cpool.map(_onehot4, range(8), chunksize=2, reduce="sum",
                        reduce_spec=((4,), torch.float32))
        assert torch.equal(red, torch.full((4,), float(sum(range(8)))))


def _use_theta(x, theta=None):
    return float(theta[0]) + x


class TestTransferReduction:
    def test_shared_ships_once_not_per_chunk(self, cpool, monkeypatch):
        """VERDICT r1 criterion: >=10x fewer tensor transfers than the
        per-chunk path.  Counts the actual bytes the master serializes
        for dispatch: a functools.partial tensor rides in EVERY chunk;
        shared= ships ONE blob (to rank 0) + tiny chunk payloads."""
        import functools

        import fiber_amd.pool as pm

        theta = torch.ones(100_000)  # 400 KB
        counts = {"bytes": 0}
        real_dumps = pm.serialization.dumps

        def counting_dumps(obj, *a, **k):
            blob = real_dumps(obj, *a, **k)
            counts["bytes"] += len(blob)
            return blob

        monkeypatch.setattr(pm.serialization, "dumps", counting_dumps)

        counts["bytes"] = 0
        cpool.map(functools.partial(_use_theta, theta=theta), range(32),
                  chunksize=1)
        per_chunk_bytes = counts["bytes"]

        counts["bytes"] = 0
        cpool.map(_use_theta, range(32), chunksize=1,
                  shared={"theta": theta})
        shared_bytes = counts["bytes"]

        assert per_chunk_bytes > 32 * 390_000  # tensor in every chunk
        # one staging blob (serialized twice: build + rank-0 ctl record)
        # + tiny chunk payloads
        assert shared_bytes < 1_000_000
        assert per_chunk_bytes / shared_bytes >= 10
