"""Observability: structured logging + per-phase timers."""
from __future__ import annotations

import contextlib
import logging
import time

_log = logging.getLogger("lfmq")
if not _log.handlers:
    h = logging.StreamHandler()
    h.setFormatter(logging.Formatter("[lfmq %(asctime)s] %(message)s",
                                     datefmt="%H:%M:%S"))
    _log.addHandler(h)
    _log.setLevel(logging.INFO)


def get_logger() -> logging.Logger:
    return _log


@contextlib.contextmanager
def phase_timer(name: str):
    t0 = time.perf_counter()
    _log.info("%s ...", name)
    try:
        yield
    finally:
        _log.info("%s done in %.2fs", name, time.perf_counter() - t0)
