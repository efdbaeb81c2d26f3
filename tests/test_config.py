"""Config tests (parity: reference tests/test_config.py)."""

import os

import pytest

import fiber_amd
from fiber_amd import config as fam_config
from fiber_amd.queues import SimpleQueue


def _report_config(q):
    conf = fam_config.get_object()
    q.put((conf.log_level, conf.cpu_per_job))


class TestConfig:
    def test_defaults(self):
        conf = fam_config.Config()
        assert conf.default_backend == "local"
        assert conf.cpu_per_job == 1

    def test_file_parse(self, tmp_path):
        path = tmp_path / "famconfig"
        path.write_text("[default]\nlog_level=debug\ncpu_per_job=3\n")
        conf = fam_config.Config(conf_file=str(path))
        assert conf.log_level == "debug"
        assert conf.cpu_per_job == 3

    def test_invalid_key_raises(self, tmp_path):
        path = tmp_path / "famconfig"
        path.write_text("[default]\nbogus_key=1\n")
        with pytest.raises(ValueError):
            fam_config.Config(conf_file=str(path))

    def test_invalid_kwarg_raises(self):
        with pytest.raises(ValueError):
            fam_config.Config(not_a_key=1)

    def test_env_override(self, monkeypatch):
        monkeypatch.setenv("FAM_LOG_LEVEL", "warning")
        conf = fam_config.Config()
        assert conf.log_level == "warning"

    def test_kwargs_beat_env(self, monkeypatch):
        monkeypatch.setenv("FAM_LOG_LEVEL", "warning")
        conf = fam_config.Config(log_level="error")
        assert conf.log_level == "error"

    def test_module_globals_mirror(self):
        fam_config.init(log_level="debug")
        try:
            assert fam_config.log_level == "debug"
        finally:
            fam_config.init()

    def test_config_syncs_to_child(self):
        """Children inherit the master's exact config (reference
        test_config.py:45-55)."""
        fam_config.init(log_level="warning", cpu_per_job=2)
        try:
            q = SimpleQueue()
            p = fiber_amd.Process(target=_report_config, args=(q,))
            p.start()
            assert q.get(timeout=30) == ("warning", 2)
            p.join(30)
            q.close()
        finally:
            fam_config.init()

    def test_bool_coercion(self):
        conf = fam_config.Config(debug="true")
        assert conf.debug is True
        conf = fam_config.Config(debug="0")
        assert conf.debug is False
