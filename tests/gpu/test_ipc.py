"""HIP IPC data plane: CUDA tensors cross process boundaries as handles,
never by value (the reference's nanomsg byte plane carried pickled bytes;
here device tensors stay device-resident — SURVEY §2c)."""

import pytest
import torch

import fiber_amd
from fiber_amd.queues import SimpleQueue

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(
    not torch.cuda.is_available(), reason="needs MI355X"
)


def _consumer(q, out):
    t = q.get(timeout=60)
    assert t.is_cuda, "tensor arrived as CPU copy — IPC path broken"
    out.put(float(t.sum().item()))
    # in-place mutation must be visible to the producer (shared storage)
    t.fill_(7.0)
    torch.cuda.synchronize()
    out.put("mutated")


@requires_gpu
class TestHipIpcTensors:
    def test_tensor_via_queue_zero_copy(self):
        q, out = SimpleQueue(), SimpleQueue()
        t = torch.arange(1024, dtype=torch.float32, device="cuda")
        p = fiber_amd.Process(target=_consumer, args=(q, out))
        p.start()
        q.put(t)
        assert out.get(timeout=120) == float(t.numel() * (t.numel() - 1) / 2)
        assert out.get(timeout=60) == "mutated"
        p.join(60)
        assert p.exitcode == 0
        torch.cuda.synchronize()
        # shared storage: the child's fill_ is visible here
        assert torch.all(t == 7.0)
        q.close()
        out.close()

    def test_pool_map_with_device_args(self):
        from fiber_amd.pool import ZPool

        pool = ZPool(processes=2, gpu_per_worker=1)
        try:
            tensors = [
                torch.full((64,), float(i), device="cuda") for i in range(8)
            ]
            res = pool.map(_tensor_sum, tensors)
            assert res == [float(i) * 64 for i in range(8)]
        finally:
            pool.terminate()
            pool.join()


def _tensor_sum(t):
    assert t.is_cuda
    return float(t.sum().item())


def _batch_consumer(q, out):
    got = []
    while len(got) < 8:
        got.extend(q.get_many(max_n=8, timeout=60))
    assert all(t.is_cuda for t in got)
    out.put([float(t.sum().item()) for t in got])


@requires_gpu
class TestBatchedIpcTensors:
    def test_put_many_cuda_tensors(self):
        """Batched queue ops compose with the HIP-IPC tensor reducers:
        each batched message still carries only an IPC handle."""
        q, out = SimpleQueue(), SimpleQueue()
        tensors = [
            torch.full((256,), float(i), device="cuda") for i in range(8)
        ]
        torch.cuda.synchronize()
        p = fiber_amd.Process(target=_batch_consumer, args=(q, out))
        p.start()
        q.put_many(tensors)
        sums = out.get(timeout=120)
        assert sums == [256.0 * i for i in range(8)]
        p.join(60)
        assert p.exitcode == 0
        q.close()
        out.close()
