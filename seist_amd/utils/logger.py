"""Process-wide logger singleton.

Capability parity with the reference's named-logger singleton
(/root/reference/utils/logger.py:5-89): one shared log directory, named
loggers ("global"/"train"/"test") each writing ``<name>.log`` plus stdout,
and attribute proxying so ``logger.info(...)`` hits the active logger.
"""

import logging
import os
import sys
from typing import Optional

_FMT = "%(asctime)s [%(levelname)s] %(message)s"
_DATEFMT = "%Y-%m-%d %H:%M:%S"


class _Logger:
    def __init__(self) -> None:
        self._logdir: Optional[str] = None
        self._loggers = {}
        self._active = "global"
        self._ensure("global")

    def _ensure(self, name: str) -> logging.Logger:
        if name not in self._loggers:
            lg = logging.getLogger(f"seist_amd.{name}")
            lg.setLevel(logging.INFO)
            lg.propagate = False
            sh = logging.StreamHandler(sys.stdout)
            sh.setFormatter(logging.Formatter(_FMT, _DATEFMT))
            lg.addHandler(sh)
            self._loggers[name] = lg
            if self._logdir is not None:
                self._attach_file(name)
        return self._loggers[name]

    def _attach_file(self, name: str) -> None:
        lg = self._loggers[name]
        path = os.path.join(self._logdir, f"{name}.log")
        if not any(
            isinstance(h, logging.FileHandler)
            and getattr(h, "baseFilename", None) == os.path.abspath(path)
            for h in lg.handlers
        ):
            fh = logging.FileHandler(path)
            fh.setFormatter(logging.Formatter(_FMT, _DATEFMT))
            lg.addHandler(fh)

    def set_logdir(self, logdir: str) -> None:
        os.makedirs(logdir, exist_ok=True)
        self._logdir = logdir
        for name in self._loggers:
            self._attach_file(name)

    def get_logdir(self) -> Optional[str]:
        return self._logdir

    def logdir(self) -> Optional[str]:
        return self._logdir

    def set_logger(self, name: str) -> None:
        self._ensure(name)
        self._active = name

    def __getattr__(self, attr):
        return getattr(self._loggers[self._active], attr)


logger = _Logger()
