"""Small shared utilities (logging setup, interactive-shell detection)."""

import logging
import os
import sys
import uuid

_LEVELS = {
    "debug": logging.DEBUG,
    "info": logging.INFO,
    "warning": logging.WARNING,
    "error": logging.ERROR,
    "critical": logging.CRITICAL,
}


def init_logger(config, proc_name="MainProcess"):
    """Per-process logger writing to ``log_file.<procname>`` (or stdout).

    Parity with reference ``fiber/init.py:25-49``.
    """
    logger = logging.getLogger("fiber_amd")
    logger.handlers = []
    level = _LEVELS.get(str(config.log_level).lower(), logging.INFO)
    logger.setLevel(level)
    if config.log_file == "stdout" or config.merge_output:
        handler = logging.StreamHandler(sys.stdout)
    else:
        path = "%s.%s" % (config.log_file, proc_name)
        try:
            handler = logging.FileHandler(path)
        except OSError:
            handler = logging.StreamHandler(sys.stderr)
    handler.setFormatter(
        logging.Formatter(
            "%(asctime)s %(name)s %(processName)s %(levelname)s %(message)s"
        )
    )
    logger.addHandler(handler)
    return logger


def get_logger():
    return logging.getLogger("fiber_amd")


def is_in_interactive_console():
    """True in a REPL/notebook/stdin script (closures ride cloudpickle)."""
    main = sys.modules.get("__main__")
    if not hasattr(main, "__file__"):
        return True
    name = os.path.basename(str(main.__file__))
    return name.startswith("<")  # '<stdin>', '<string>', ...


def random_name(prefix):
    return "%s-%s" % (prefix, uuid.uuid4().hex[:12])


def ipc_path(config, name):
    return os.path.join(config.ipc_dir, name)
