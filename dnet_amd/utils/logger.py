"""Singleton ``dnet`` logger with per-process file handlers and a
``[PROFILE]`` line filter gated by observability settings.

Reference counterpart: src/dnet/utils/logger.py.
"""
from __future__ import annotations

import logging
import os
import sys
from pathlib import Path

_configured = False


class ProfileLogFilter(logging.Filter):
    def __init__(self, enabled: bool):
        super().__init__()
        self.enabled = enabled

    def filter(self, record: logging.LogRecord) -> bool:
        if "[PROFILE]" in record.getMessage():
            return self.enabled
        return True


def get_logger(role: str = "dnet") -> logging.Logger:
    global _configured
    log = logging.getLogger("dnet")
    if _configured:
        return log
    from ..config import get_settings
    s = get_settings()
    log.setLevel(getattr(logging, s.logging.level.upper(), logging.INFO))
    fmt = logging.Formatter(
        "%(asctime)s %(levelname)s [%(name)s] %(message)s")
    sh = logging.StreamHandler(sys.stderr)
    sh.setFormatter(fmt)
    log.addHandler(sh)
    try:
        d = Path(os.path.expanduser(s.logging.dir))
        d.mkdir(parents=True, exist_ok=True)
        fh = logging.FileHandler(d / f"dnet-{role}-{os.getpid()}.log")
        fh.setFormatter(fmt)
        log.addHandler(fh)
    except OSError:
        pass
    log.addFilter(ProfileLogFilter(s.observability.profile))
    log.propagate = False
    _configured = True
    return log


logger = logging.getLogger("dnet")
