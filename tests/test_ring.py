"""Ring tests over gloo (world_size 2, CPU).  RCCL paths are covered by
the gpu-marked suite."""

import functools

import torch

import fiber_amd
from fiber_amd.queues import SimpleQueue
from fiber_amd.ring import Ring, RingContext, RingNode


def _noop_rank(rank, size):
    return None


def _allreduce_work(rank, size, out=None):
    ctx = RingContext(rank, size, backend="gloo")
    t = torch.ones(100) * (rank + 1)
    ctx.allreduce(t)
    out.put((rank, float(t[0])))
    ctx.shutdown()


def _grad_work(rank, size, out=None):
    ctx = RingContext(rank, size, backend="gloo", bucket_mb=1)
    params = [torch.nn.Parameter(torch.ones(3000)) for _ in range(4)]
    for p in params:
        p.grad = torch.full_like(p.data, float(rank))
    ctx.allreduce_grads(params, average=True)
    out.put((rank, [float(p.grad[0]) for p in params]))
    ctx.shutdown()


def _gather_work(rank, size, out=None):
    ctx = RingContext(rank, size, backend="gloo")
    t = torch.tensor([float(rank)])
    gathered = ctx.all_gather(t)
    ctx.barrier()
    out.put((rank, [float(g[0]) for g in gathered]))
    ctx.shutdown()


def _initializer(ctx):
    ctx.init()


class TestRing:
    def test_allreduce(self):
        out = SimpleQueue()
        ring = Ring(
            2,
            functools.partial(_allreduce_work, out=out),
            backend="gloo",
            gpu_per_rank=0,
        )
        ring.run(timeout=120)
        vals = sorted(out.get(timeout=10) for _ in range(2))
        assert vals == [(0, 3.0), (1, 3.0)]
        out.close()

    def test_bucketed_grad_allreduce(self):
        out = SimpleQueue()
        ring = Ring(
            2,
            functools.partial(_grad_work, out=out),
            backend="gloo",
            gpu_per_rank=0,
        )
        ring.run(timeout=120)
        for _ in range(2):
            _rank, grads = out.get(timeout=10)
            assert grads == [0.5] * 4
        out.close()

    def test_all_gather_and_barrier(self):
        out = SimpleQueue()
        ring = Ring(
            2,
            functools.partial(_gather_work, out=out),
            initializer=_initializer,
            backend="gloo",
            gpu_per_rank=0,
        )
        ring.run(timeout=120)
        for _ in range(2):
            _rank, gathered = out.get(timeout=10)
            assert gathered == [0.0, 1.0]
        out.close()

    def test_members_table(self):
        ring = Ring(3, lambda r, s: None, gpu_per_rank=0)
        assert [m.rank for m in ring.members] == [0, 1, 2]
        assert isinstance(ring.members[0], RingNode)

    def test_membership_from_real_child_feedback(self):
        """Every member's pid comes from the rank's own 'up' report —
        not optimistic bookkeeping.  After a clean
        run the nodes report done and are marked disconnected."""
        ring = Ring(2, _noop_rank, backend="gloo", gpu_per_rank=0)
        ring.run(timeout=300)
        pids = [m.pid for m in ring.members]
        assert all(isinstance(p, int) and p > 0 for p in pids)
        assert len(set(pids)) == 2
        assert all(m.connected is False for m in ring.members)

    def test_failed_rank_raises(self):
        def boom(rank, size):
            if rank == 1:
                raise RuntimeError("rank down")

        ring = Ring(2, boom, backend="gloo", gpu_per_rank=0)
        try:
            ring.run(timeout=120)
            raised = False
        except RuntimeError:
            raised = True
        assert raised


class TestExperimentalAlias:
    def test_reference_import_path(self):
        from fiber_amd.experimental.ring import Ring as R2
        from fiber_amd.experimental.ring import RingNode as N2

        assert R2 is Ring
        assert N2 is RingNode
