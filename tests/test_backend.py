"""Backend registry tests (parity: reference tests/test_backend.py)."""

import pytest

from fiber_amd import backend as fam_backend
from fiber_amd import config as fam_config
from fiber_amd.backends.local import Backend as LocalBackend
from fiber_amd.core import JobSpec, ProcessStatus


class TestRegistry:
    def test_default_is_local(self):
        assert fam_backend.auto_select_backend() == "local"
        backend = fam_backend.get_backend()
        assert isinstance(backend, LocalBackend)

    def test_singleton(self):
        assert fam_backend.get_backend("local") is fam_backend.get_backend(
            "local"
        )

    def test_unknown_backend_raises(self):
        with pytest.raises(ValueError):
            fam_backend.get_backend("kubernetes")

    def test_config_selects_backend(self):
        fam_config.init(backend="local")
        try:
            assert fam_backend.auto_select_backend() == "local"
        finally:
            fam_config.init()

    def test_hot_swap_seam(self):
        class Fake(LocalBackend):
            name = "fake"

        fam_backend.set_backend("fake", Fake())
        try:
            assert isinstance(fam_backend.get_backend("fake"), Fake)
        finally:
            fam_backend._backends.pop("fake", None)
            fam_backend.available_backend.remove("fake")


class TestLocalBackend:
    def test_job_lifecycle(self):
        backend = LocalBackend()
        spec = JobSpec(command=["sleep", "30"], name="t")
        job = backend.create_job(spec)
        assert backend.get_job_status(job) == ProcessStatus.STARTED
        backend.terminate_job(job)
        code = backend.wait_for_job(job, 10)
        assert code is not None
        assert backend.get_job_status(job) == ProcessStatus.STOPPED

    def test_device_pinning_env(self):
        backend = LocalBackend()
        spec = JobSpec(
            command=["python", "-c", "import os,sys;sys.exit(0 if os.environ.get('HIP_VISIBLE_DEVICES') is not None else 3)"],
            gpu=1,
        )
        job = backend.create_job(spec)
        code = backend.wait_for_job(job, 30)
        assert code == 0
        assert job.devices is not None and len(job.devices) == 1

    def test_round_robin_device_assignment(self):
        backend = LocalBackend()
        first = backend._assign_devices(1)
        second = backend._assign_devices(1)
        assert len(first) == len(second) == 1
