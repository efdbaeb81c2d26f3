"""RCCL path on hardware (world 1 — single-GPU boxes; the multi-GPU
scaling run is the driver's).  Exercises the exact collective code path
bench.py uses at N>1."""

import os

import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(
    not torch.cuda.is_available(), reason="needs MI355X"
)


@requires_gpu
class TestRcclWorldOne:
    def _ctx(self):
        from fiber_amd.ring import RingContext, _free_tcp_port

        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(_free_tcp_port())
        os.environ.setdefault("RANK", "0")
        os.environ.setdefault("WORLD_SIZE", "1")
        return RingContext(0, 1, backend="nccl",
                           device=torch.device("cuda", 0))

    def test_collectives_and_es_step(self):
        ctx = self._ctx()
        try:
            ctx.init()
            t = torch.ones(1024, device="cuda")
            ctx.allreduce(t)
            assert torch.all(t == 1.0)
            out = torch.empty(1024, device="cuda")
            ctx.all_gather_into(out, t)
            assert torch.all(out == 1.0)
            ctx.barrier()

            # the distributed ES step end-to-end over RCCL (world 1)
            from fiber_amd.es import ESConfig, ESEngine

            engine = ESEngine(ESConfig(pop_per_gpu=64, horizon=8), ctx=ctx,
                              device=torch.device("cuda", 0))
            stats = engine.step()
            torch.cuda.synchronize()
            assert stats["grad_norm"] > 0
        finally:
            ctx.shutdown()
