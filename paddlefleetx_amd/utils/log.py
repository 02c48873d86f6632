"""Singleton color logger with TRAIN/EVAL levels and the `ips:` log grammar.

Reference surface: ppfleetx/utils/log.py:33-150 (custom levels, color),
language_module.py:108-113 (ips log line the benchmark harness parses).
"""

from __future__ import annotations

import logging
import os
import sys

__all__ = ["logger"]

TRAIN = 21
EVAL = 22
IMPORTANT = 23
logging.addLevelName(TRAIN, "TRAIN")
logging.addLevelName(EVAL, "EVAL")
logging.addLevelName(IMPORTANT, "IMPORTANT")

_COLORS = {
    "DEBUG": "\033[37m",
    "INFO": "\033[36m",
    "TRAIN": "\033[32m",
    "EVAL": "\033[33m",
    "IMPORTANT": "\033[35m",
    "WARNING": "\033[33m",
    "ERROR": "\033[31m",
}
_RESET = "\033[0m"


class _Formatter(logging.Formatter):
    def format(self, record):
        msg = super().format(record)
        if sys.stdout.isatty():
            color = _COLORS.get(record.levelname, "")
            return f"{color}{msg}{_RESET}" if color else msg
        return msg


class _Logger(logging.Logger):
    def train(self, msg, *args, **kwargs):
        if self.isEnabledFor(TRAIN):
            self._log(TRAIN, msg, args, **kwargs)

    def eval(self, msg, *args, **kwargs):
        if self.isEnabledFor(EVAL):
            self._log(EVAL, msg, args, **kwargs)

    def important(self, msg, *args, **kwargs):
        if self.isEnabledFor(IMPORTANT):
            self._log(IMPORTANT, msg, args, **kwargs)


def _build_logger() -> _Logger:
    logging.setLoggerClass(_Logger)
    lg = logging.getLogger("paddlefleetx_amd")
    logging.setLoggerClass(logging.Logger)
    rank = int(os.environ.get("RANK", "0"))
    lg.setLevel(logging.INFO if rank == 0 else logging.WARNING)
    if not lg.handlers:
        h = logging.StreamHandler(sys.stdout)
        h.setFormatter(_Formatter("[%(asctime)s] [%(levelname)s] %(message)s",
                                  datefmt="%Y-%m-%d %H:%M:%S"))
        lg.addHandler(h)
    lg.propagate = False
    return lg


logger: _Logger = _build_logger()
