"""Component logging in the reference's format: "ts LEVEL: file:line msg"
into /kubeshare/log/<component>.log + stderr (reference
pkg/logger/logger.go:40-57; the hostPath is shared with the native
daemons' logs)."""
from __future__ import annotations

import logging
import os
import sys

from . import constants as C

_FMT = "%(asctime)s.%(msecs)03d %(levelname)s: %(filename)s:%(lineno)d %(message)s"
_DATEFMT = "%Y-%m-%d %H:%M:%S"


def get_logger(component: str, log_dir: str | None = None,
               level: int = logging.INFO) -> logging.Logger:
    logger = logging.getLogger(f"kubeshare.{component}")
    if logger.handlers:
        return logger
    logger.setLevel(level)
    fmt = logging.Formatter(_FMT, _DATEFMT)
    sh = logging.StreamHandler(sys.stderr)
    sh.setFormatter(fmt)
    logger.addHandler(sh)
    log_dir = log_dir or os.environ.get("KUBESHARE_LOG_DIR", C.LOG_PATH)
    try:
        os.makedirs(log_dir, exist_ok=True)
        fh = logging.FileHandler(os.path.join(log_dir, f"{component}.log"))
        fh.setFormatter(fmt)
        logger.addHandler(fh)
    except OSError:
        pass  # read-only container without the hostPath: stderr only
    return logger
