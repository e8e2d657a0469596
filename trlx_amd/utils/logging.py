"""Library-wide logging with multi-process rank filtering.

Parity target: reference trlx/utils/logging.py — ``TRLX_VERBOSITY`` env
control, ``MultiProcessAdapter`` with ``ranks=[...]`` filtering and
``[RANK n]`` prefixes, tqdm on/off wrappers.
"""

import logging
import os
import sys
import threading
from logging import (  # noqa: F401
    CRITICAL,
    DEBUG,
    ERROR,
    FATAL,
    INFO,
    NOTSET,
    WARNING,
)
from typing import Optional

from tqdm import auto as tqdm_lib

_lock = threading.Lock()
_default_handler: Optional[logging.Handler] = None

log_levels = {
    "debug": DEBUG,
    "info": INFO,
    "warning": WARNING,
    "error": ERROR,
    "critical": CRITICAL,
}

_default_log_level = INFO


def _get_default_logging_level():
    env_level_str = os.getenv("TRLX_VERBOSITY", None)
    if env_level_str:
        if env_level_str.lower() in log_levels:
            return log_levels[env_level_str.lower()]
        logging.getLogger().warning(
            f"Unknown option TRLX_VERBOSITY={env_level_str}, has to be one of: {', '.join(log_levels.keys())}"
        )
    return _default_log_level


def _get_library_name() -> str:
    return __name__.split(".")[0]


def _get_library_root_logger() -> logging.Logger:
    return logging.getLogger(_get_library_name())


def _configure_library_root_logger() -> None:
    global _default_handler
    with _lock:
        if _default_handler:
            return
        _default_handler = logging.StreamHandler()  # sys.stderr as stream
        _default_handler.flush = sys.stderr.flush

        formatter = logging.Formatter(
            "[%(asctime)s] [%(levelname)s] [%(name)s] %(message)s",
            datefmt="%Y-%m-%d %H:%M:%S",
        )
        _default_handler.setFormatter(formatter)

        library_root_logger = _get_library_root_logger()
        library_root_logger.addHandler(_default_handler)
        library_root_logger.setLevel(_get_default_logging_level())
        library_root_logger.propagate = False


class MultiProcessAdapter(logging.LoggerAdapter):
    """A logger adapter for distributed runs.

    ``logger.info(msg, ranks=[0])`` logs only on the listed ranks; by default
    every rank logs with a ``[RANK n]`` prefix (rank 0 unprefixed).
    """

    @staticmethod
    def _rank() -> int:
        return int(os.environ.get("RANK", 0))

    def log(self, level, msg, *args, **kwargs):
        ranks = kwargs.pop("ranks", None)
        rank = self._rank()
        if ranks is None or rank in ranks or -1 in ranks:
            if self.isEnabledFor(level):
                if rank != 0:
                    msg = f"[RANK {rank}] {msg}"
                msg, kwargs = self.process(msg, kwargs)
                self.logger.log(level, msg, *args, **kwargs)

    def process(self, msg, kwargs):
        # strip our custom kwarg before stdlib sees it
        kwargs.pop("ranks", None)
        return msg, kwargs


def get_logger(name: Optional[str] = None) -> MultiProcessAdapter:
    """Return the library logger (rank-aware)."""
    if name is None:
        name = _get_library_name()
    _configure_library_root_logger()
    logger = logging.getLogger(name)
    return MultiProcessAdapter(logger, {})


def get_verbosity() -> int:
    _configure_library_root_logger()
    return _get_library_root_logger().getEffectiveLevel()


def set_verbosity(verbosity: int) -> None:
    _configure_library_root_logger()
    _get_library_root_logger().setLevel(verbosity)


def set_verbosity_debug():
    set_verbosity(DEBUG)


def set_verbosity_info():
    set_verbosity(INFO)


def set_verbosity_warning():
    set_verbosity(WARNING)


def set_verbosity_error():
    set_verbosity(ERROR)


def disable_default_handler() -> None:
    _configure_library_root_logger()
    assert _default_handler is not None
    _get_library_root_logger().removeHandler(_default_handler)


def enable_default_handler() -> None:
    _configure_library_root_logger()
    assert _default_handler is not None
    _get_library_root_logger().addHandler(_default_handler)


def add_handler(handler: logging.Handler) -> None:
    _configure_library_root_logger()
    assert handler is not None
    _get_library_root_logger().addHandler(handler)


def remove_handler(handler: logging.Handler) -> None:
    _configure_library_root_logger()
    assert handler is not None and handler in _get_library_root_logger().handlers
    _get_library_root_logger().removeHandler(handler)


def disable_propagation() -> None:
    _configure_library_root_logger()
    _get_library_root_logger().propagate = False


def enable_propagation() -> None:
    _configure_library_root_logger()
    _get_library_root_logger().propagate = True


# --- tqdm control ----------------------------------------------------------

_tqdm_active = True


class EmptyTqdm:
    """Dummy tqdm that doesn't do anything."""

    def __init__(self, *args, **kwargs):
        self._iterator = args[0] if args else None

    def __iter__(self):
        return iter(self._iterator)

    def __getattr__(self, _):
        def empty_fn(*args, **kwargs):
            return

        return empty_fn

    def __enter__(self):
        return self

    def __exit__(self, type_, value, traceback):
        return


class _tqdm_cls:
    def __call__(self, *args, **kwargs):
        if _tqdm_active:
            return tqdm_lib.tqdm(*args, **kwargs)
        return EmptyTqdm(*args, **kwargs)

    def set_lock(self, *args, **kwargs):
        self._lock = None
        if _tqdm_active:
            return tqdm_lib.tqdm.set_lock(*args, **kwargs)

    def get_lock(self):
        if _tqdm_active:
            return tqdm_lib.tqdm.get_lock()


tqdm = _tqdm_cls()


def is_progress_bar_enabled() -> bool:
    return bool(_tqdm_active)


def enable_progress_bar():
    global _tqdm_active
    _tqdm_active = True


def disable_progress_bar():
    global _tqdm_active
    _tqdm_active = False
