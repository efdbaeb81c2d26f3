"""Three-source configuration system.

Precedence (lowest to highest): ``.famconfig`` INI file < ``FAM_*``
environment variables < Python keyword arguments passed to
:func:`fiber_amd.init` / :func:`init`.

Design parity with the reference config system (uber/fiber
``fiber/config.py:15-65,92-249``): a ``Config`` dataclass-like object whose
fields mirror module-level globals, an ``init()`` that re-reads all three
sources, and strict rejection of unknown keys.  The key set is MI355X-native:
the docker/kubernetes image keys are gone; GPU-placement keys are added.
"""

import configparser
import os

_CONFIG_FILE = ".famconfig"
_ENV_PREFIX = "FAM_"

# (name, type, default)
_FIELDS = [
    ("debug", bool, False),
    ("backend", str, None),          # explicit backend name
    ("default_backend", str, "local"),
    ("log_level", str, "info"),
    ("log_file", str, "/tmp/fiber_amd.log"),
    ("merge_output", bool, False),
    ("cpu_per_job", int, 1),
    ("gpu_per_job", int, 0),
    # Comma-separated list of device ordinals this master may hand out.
    # Empty string = all visible devices.
    ("devices", str, ""),
    # Directory for admin unix sockets and shm ring bookkeeping files.
    ("ipc_dir", str, "/tmp"),
    # Default byte capacity of a shm ring (per direction).
    ("ring_capacity", int, 8 << 20),
    # Max in-flight tasks before the pool's task feeder throttles.
    ("max_inflight", int, 20000),
    # Seconds to wait for a worker process to dial back before failing start.
    ("start_timeout", float, 60.0),
]

_FIELD_NAMES = {f[0] for f in _FIELDS}


def _coerce(typ, value):
    if isinstance(value, str):
        if typ is bool:
            return value.strip().lower() in ("1", "true", "yes", "on")
        return typ(value)
    return typ(value) if value is not None else None


class Config:
    """Holds the effective configuration.

    ``Config(conf_file=None, **kwargs)`` reads the INI file (if present),
    then environment variables, then applies ``kwargs``.
    """

    def __init__(self, conf_file=None, **kwargs):
        for name, _typ, default in _FIELDS:
            setattr(self, name, default)

        path = conf_file or _CONFIG_FILE
        if os.path.exists(path):
            parser = configparser.ConfigParser()
            parser.read(path)
            if parser.has_section("default"):
                for key, value in parser.items("default"):
                    self._set_checked(key, value, source=path)

        for name, typ, _default in _FIELDS:
            env_key = _ENV_PREFIX + name.upper()
            if env_key in os.environ:
                setattr(self, name, _coerce(typ, os.environ[env_key]))

        for key, value in kwargs.items():
            if value is None:
                continue
            self._set_checked(key, value, source="kwargs")

    def _set_checked(self, key, value, source):
        if key not in _FIELD_NAMES:
            raise ValueError(
                "invalid config key %r (from %s); valid keys: %s"
                % (key, source, sorted(_FIELD_NAMES))
            )
        typ = next(f[1] for f in _FIELDS if f[0] == key)
        setattr(self, key, _coerce(typ, value))

    def get_dict(self):
        return {name: getattr(self, name) for name, _t, _d in _FIELDS}

    def __repr__(self):
        return "Config(%s)" % ", ".join(
            "%s=%r" % (k, v) for k, v in sorted(self.get_dict().items())
        )


# Module-level mirror of the current Config (reference parity:
# fiber/config.py:221-249 keeps module globals in sync so that
# ``fiber_amd.config.log_level`` etc. read naturally).
_current = Config()


def init(conf_file=None, **kwargs):
    """Re-initialize global config from all three sources."""
    global _current
    _current = Config(conf_file=conf_file, **kwargs)
    globals().update(_current.get_dict())
    return _current


def get_object():
    return _current


def get_dict():
    return _current.get_dict()


# populate module globals at import
globals().update(_current.get_dict())
