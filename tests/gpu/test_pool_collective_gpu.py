"""GPU tests for the pool RCCL data plane (world 1 on the single-GPU CI
box; the world-2 choreography is pinned by the CPU/gloo suite and the
driver's 8-GPU scaling run).

These verify the device path end-to-end: a CUDA tensor map-arg crosses
as ONE HIP IPC hand-off + device broadcast, reduce='sum' fans in on
device, and the flagship ES engine steps inside a GPU-pinned pool
worker with real HIP kernels (native extension, no fallback)."""

import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(
    not torch.cuda.is_available(), reason="needs MI355X"
)


def _dot_with_shared(x, theta=None):
    assert theta.is_cuda, "shared tensor must stay device-resident"
    return float(theta.sum()) * x


def _gpu_onehot(x):
    t = torch.zeros(8, device="cuda")
    t[x % 8] = float(x)
    return t


def _slow_echo_gpu(x, theta=None):
    import time

    time.sleep(0.2)
    return x + float(theta[0])


def _gpu_group_probe():
    from fiber_amd.pool import current_worker_group

    g = current_worker_group()
    t = torch.ones(4, device="cuda")
    g.allreduce(t)  # world-1 RCCL collective on the NEW generation
    return (g.rank, g.size, bool((t == 1).all().item()))


def _engine_probe(k):
    from examples.es_pool import es_steps

    return es_steps(0, k, True)


@requires_gpu
class TestPoolCollectiveGPU:
    def test_shared_cuda_tensor_broadcast(self):
        from fiber_amd.pool import ZPool

        theta = torch.arange(1024, dtype=torch.float32, device="cuda")
        pool = ZPool(processes=1, gpu_per_worker=1, collective=True)
        try:
            out = pool.map(_dot_with_shared, range(6), chunksize=1,
                           shared={"theta": theta})
            want = [float(theta.sum()) * x for x in range(6)]
            assert out == want
        finally:
            pool.terminate()
            pool.join()

    def test_shared_cpu_tensor_staged_pinned(self):
        """A HOST tensor in shared= stages via the pinned side-stream
        path (hipMemcpyAsync) and arrives device-resident."""
        from fiber_amd.pool import ZPool

        theta = torch.arange(4096, dtype=torch.float32)  # CPU
        pool = ZPool(processes=1, gpu_per_worker=1, collective=True)
        try:
            out = pool.map(_dot_with_shared, range(4), chunksize=1,
                           shared={"theta": theta})
            want = [float(theta.sum()) * x for x in range(4)]
            assert out == want
        finally:
            pool.terminate()
            pool.join()

    def test_reduce_sum_on_device(self):
        from fiber_amd.pool import ZPool

        pool = ZPool(processes=1, gpu_per_worker=1, collective=True)
        try:
            got = pool.map(_gpu_onehot, range(16), chunksize=4,
                           reduce="sum",
                           reduce_spec=((8,), torch.float32))
            want = torch.zeros(8)
            for x in range(16):
                want[x % 8] += float(x)
            assert torch.equal(got.cpu(), want)
        finally:
            pool.terminate()
            pool.join()

    def test_kill_recover_world1_nccl(self, monkeypatch):
        """The generation-rotation recovery on the REAL RCCL backend:
        kill the GPU worker mid collective map, the map fails with
        diagnosis, the respawned worker re-initializes a fresh NCCL
        generation and SPMD work succeeds."""
        import time

        from fiber_amd.pool import ZPool

        monkeypatch.setenv("FAM_PG_TIMEOUT", "30")
        pool = ZPool(processes=1, gpu_per_worker=1, collective=True)
        try:
            theta = torch.ones(8, device="cuda")
            res = pool.map_async(_slow_echo_gpu, range(40), chunksize=1,
                                 shared={"theta": theta})
            time.sleep(3.0)  # let the worker stage + start chunks
            with pool._worker_lock:
                victim = next(iter(pool._workers.values()))
            victim.kill()
            with pytest.raises(RuntimeError, match="collective"):
                res.get(120)
            deadline = time.monotonic() + 180
            while True:
                try:
                    out = pool.run_on_all(_gpu_group_probe, timeout=90)
                    break
                except Exception:
                    if time.monotonic() > deadline:
                        raise
                    time.sleep(1.0)
            assert out == [(0, 1, True)]
        finally:
            pool.terminate()
            pool.join()

    def test_es_engine_through_pool(self):
        """The flagship engine steps inside a GPU-pinned pool worker
        (real HIP rollout kernels; fail-loud if the extension is
        missing)."""
        from fiber_amd.pool import ZPool
        from examples.es_pool import init_es_worker

        cfg = dict(pop_per_gpu=512, horizon=32)
        pool = ZPool(processes=1, gpu_per_worker=1, collective=True,
                     initializer=init_es_worker, initargs=(cfg, "mlp"))
        try:
            stats = pool.run_on_all(_engine_probe, (2,), timeout=600)[0]
            assert stats["rollouts"] == 512 * 64
            assert stats["grad_norm"] > 0.0
        finally:
            pool.terminate()
            pool.join()
