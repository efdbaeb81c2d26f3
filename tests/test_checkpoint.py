"""ES checkpoint/resume + pool observability (additive aux subsystems)."""

import torch

from fiber_amd.es import ESConfig, ESEngine
from fiber_amd.pool import ZPool


def _square(x):
    return x * x


class TestCheckpoint:
    def test_roundtrip(self, tmp_path):
        cfg = ESConfig(pop_per_gpu=8, horizon=4, seed=5)
        eng = ESEngine(cfg, ctx=None, device=torch.device("cpu"))
        eng.theta += 0.5
        eng.adam_m += 0.25
        eng.t_step = 7
        eng.obs_sum += 1.0
        eng.obs_count += 100.0
        path = str(tmp_path / "es.pt")
        eng.save(path)

        eng2 = ESEngine(cfg, ctx=None, device=torch.device("cpu"))
        assert not torch.equal(eng2.theta, eng.theta)
        eng2.load(path)
        assert torch.equal(eng2.theta, eng.theta)
        assert torch.equal(eng2.adam_m, eng.adam_m)
        assert eng2.t_step == 7
        # load recomputes obs_mu from the restored moment sums
        assert torch.allclose(eng2.obs_mu, eng.obs_sum / eng.obs_count)

    def test_state_dict_has_config(self):
        cfg = ESConfig(pop_per_gpu=8, horizon=4)
        eng = ESEngine(cfg, ctx=None, device=torch.device("cpu"))
        state = eng.state_dict()
        assert state["config"]["pop_per_gpu"] == 8


class TestPoolStats:
    def test_counters(self):
        pool = ZPool(processes=2)
        try:
            pool.map(_square, range(64))
            stats = pool.stats()
            assert stats["tasks_sent"] >= 2
            assert stats["results_received"] == stats["tasks_sent"]
            assert stats["in_flight"] == 0
            assert stats["workers_alive"] == 2
            assert stats["state"] == "run"
        finally:
            pool.terminate()
            pool.join()
