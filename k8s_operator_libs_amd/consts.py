"""Shared logging-level constants.

Capability parity with the reference's ``pkg/consts/consts.go:24-29``, which
defines logr verbosity levels following the zap convention (error levels
negative, debug levels positive).  Python callers map these onto the stdlib
``logging`` module via :func:`to_logging_level`.
"""

import logging

# zap-convention verbosity levels (reference pkg/consts/consts.go:24-29)
LOG_LEVEL_ERROR = -2
LOG_LEVEL_WARNING = -1
LOG_LEVEL_INFO = 0
LOG_LEVEL_DEBUG = 1

_LOGGING_MAP = {
    LOG_LEVEL_ERROR: logging.ERROR,
    LOG_LEVEL_WARNING: logging.WARNING,
    LOG_LEVEL_INFO: logging.INFO,
    LOG_LEVEL_DEBUG: logging.DEBUG,
}


def to_logging_level(level: int) -> int:
    """Map a zap-convention verbosity level to a stdlib ``logging`` level."""
    if level in _LOGGING_MAP:
        return _LOGGING_MAP[level]
    # Higher zap verbosity -> more detailed than DEBUG; clamp.
    return logging.DEBUG if level > 0 else logging.ERROR
