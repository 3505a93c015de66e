"""Leveled logging (parity: pkg/common/observability/logging, logr/zap V-levels).

V-level convention from the reference (`logging/const.go:20-23`):
DEFAULT=2, VERBOSE=3, DEBUG=4, TRACE=5. Mapped onto python logging levels.
"""
import logging
import os
import sys

DEFAULT = 2
VERBOSE = 3
DEBUG = 4
TRACE = 5

_V_TO_PY = {0: logging.WARNING, 1: logging.INFO, 2: logging.INFO,
            3: logging.DEBUG, 4: logging.DEBUG, 5: logging.DEBUG}

_configured = False
_verbosity = int(os.environ.get("LDS_AMD_V", "2"))


def set_verbosity(v: int) -> None:
    global _verbosity
    _verbosity = v


def verbosity() -> int:
    return _verbosity


class VLogger:
    """logr-style logger: .v(level).info(...) gated on global verbosity."""

    def __init__(self, name: str, level: int = 0):
        self._log = logging.getLogger(name)
        self._level = level

    def v(self, level: int) -> "VLogger":
        return VLogger(self._log.name, level)

    def info(self, msg: str, **kv):
        if self._level <= _verbosity:
            self._log.log(_V_TO_PY.get(self._level, logging.DEBUG),
                          _fmt(msg, kv))

    def error(self, msg: str, **kv):
        self._log.error(_fmt(msg, kv))

    def warning(self, msg: str, **kv):
        self._log.warning(_fmt(msg, kv))


def _fmt(msg, kv):
    if not kv:
        return msg
    return msg + " " + " ".join(f"{k}={v!r}" for k, v in kv.items())


def get_logger(name: str) -> VLogger:
    global _configured
    if not _configured:
        logging.basicConfig(
            stream=sys.stderr,
            level=logging.INFO,
            format="%(asctime)s %(levelname).1s %(name)s: %(message)s",
        )
        _configured = True
    return VLogger(name)
