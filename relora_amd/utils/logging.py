"""Rank-aware console logging.

The reference uses `loguru` (rank 0 only — reference torchrun_main.py:371).
loguru is not available in this image, so this module provides a compatible
`logger` object (info/warning/error/debug + remove/add) on top of stdlib
logging. Import as `from relora_amd.utils.logging import logger`.
"""

import logging
import sys

_FMT = "%(asctime)s | %(levelname)-7s | %(message)s"


class _Logger:
    def __init__(self):
        self._logger = logging.getLogger("relora_amd")
        self._logger.setLevel(logging.INFO)
        self._handler = logging.StreamHandler(sys.stderr)
        self._handler.setFormatter(logging.Formatter(_FMT, datefmt="%Y-%m-%d %H:%M:%S"))
        self._logger.addHandler(self._handler)
        self._logger.propagate = False
        self._enabled = True

    # loguru-compatible surface -------------------------------------------------
    def remove(self, *args, **kwargs):
        """Disable output (loguru's logger.remove()); used on nonzero ranks."""
        self._enabled = False

    def add(self, sink=sys.stderr, **kwargs):
        self._enabled = True

    def info(self, msg, *args, **kwargs):
        if self._enabled:
            self._logger.info(str(msg))

    def warning(self, msg, *args, **kwargs):
        if self._enabled:
            self._logger.warning(str(msg))

    def error(self, msg, *args, **kwargs):
        if self._enabled:
            self._logger.error(str(msg))

    def debug(self, msg, *args, **kwargs):
        if self._enabled:
            self._logger.debug(str(msg))

    def exception(self, msg, *args, **kwargs):
        if self._enabled:
            self._logger.exception(str(msg))


logger = _Logger()
