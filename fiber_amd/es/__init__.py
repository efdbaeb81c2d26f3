from .engine import ESConfig, ESEngine, init_theta, make_env_params  # noqa: F401
