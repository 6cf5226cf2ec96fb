"""Dedicated 'autodist' logger: stderr + timestamped file.

Reference behavior: autodist/utils/logging.py:33-146 (file under
/tmp/autodist/logs/<ts>.log + stderr handler, level from env).
"""
import datetime
import logging as _logging
import os
import sys
import threading

from autodist_amd.const import DEFAULT_LOG_DIR, ENV

_logger = None
_logger_lock = threading.Lock()


def _create_logger():
    logger = _logging.getLogger("autodist")
    logger.propagate = False
    level = ENV.AUTODIST_MIN_LOG_LEVEL.val.upper()
    logger.setLevel(getattr(_logging, level, _logging.INFO))
    fmt = _logging.Formatter(
        "%(asctime)s [autodist] %(levelname)s %(filename)s:%(lineno)d %(message)s")
    sh = _logging.StreamHandler(sys.stderr)
    sh.setFormatter(fmt)
    logger.addHandler(sh)
    try:
        os.makedirs(DEFAULT_LOG_DIR, exist_ok=True)
        ts = datetime.datetime.now().strftime("%Y%m%d-%H%M%S")
        rank = os.environ.get("RANK", "0")
        fh = _logging.FileHandler(os.path.join(DEFAULT_LOG_DIR, f"{ts}-r{rank}.log"))
        fh.setFormatter(fmt)
        logger.addHandler(fh)
    except OSError:
        pass
    return logger


def get_logger():
    global _logger
    if _logger is None:
        with _logger_lock:
            if _logger is None:
                _logger = _create_logger()
    return _logger


def debug(msg, *args, **kw):
    get_logger().debug(msg, *args, **kw, stacklevel=2)


def info(msg, *args, **kw):
    get_logger().info(msg, *args, **kw, stacklevel=2)


def warning(msg, *args, **kw):
    get_logger().warning(msg, *args, **kw, stacklevel=2)


def error(msg, *args, **kw):
    get_logger().error(msg, *args, **kw, stacklevel=2)


def set_verbosity(level):
    get_logger().setLevel(level)
