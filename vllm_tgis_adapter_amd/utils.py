"""Process-level utilities: failed-task scan, Kubernetes termination log
(reference utils.py behavior)."""

from __future__ import annotations

import asyncio
import os
from collections.abc import Iterable, Sequence
from typing import Optional


def check_for_failed_tasks(tasks: Iterable[asyncio.Task]) -> Optional[asyncio.Task]:
    for task in tasks:
        try:
            if task.exception():
                return task
        except (asyncio.InvalidStateError, asyncio.CancelledError):
            pass
    return None


def write_termination_log(msg: str, file: str = "/dev/termination-log") -> None:
    """Write the terminal failure cause where Kubernetes picks it up."""
    from .logging import DEFAULT_LOGGER_NAME, init_logger

    logger = init_logger(DEFAULT_LOGGER_NAME)
    if not os.path.exists(file):
        logger.debug("Not writing to termination log %s since it does not exist", file)
        return
    try:
        with open(file, "w") as f:
            f.write(f"{msg}\n")
    except Exception:
        logger.exception("Unable to write termination logs to %s", file)


def to_list(seq: Sequence[int]) -> list[int]:
    return seq if isinstance(seq, list) else list(seq)


class TTLCache(dict):
    """Minimal TTL-bounded dict (replaces cachetools.TTLCache, absent here)."""

    def __init__(self, maxsize: int, ttl: float):
        super().__init__()
        import time

        self._maxsize = maxsize
        self._ttl = ttl
        self._time = time.monotonic
        self._expiry: dict = {}

    def __setitem__(self, key, value):
        now = self._time()
        self._evict(now)
        if len(self) >= self._maxsize and key not in self:
            oldest = min(self._expiry, key=self._expiry.get, default=None)
            if oldest is not None:
                self.pop(oldest, None)
                self._expiry.pop(oldest, None)
        super().__setitem__(key, value)
        self._expiry[key] = now + self._ttl

    def get(self, key, default=None):
        exp = self._expiry.get(key)
        if exp is not None and exp < self._time():
            self.pop(key, None)
            self._expiry.pop(key, None)
            return default
        return super().get(key, default)

    def _evict(self, now: float) -> None:
        dead = [k for k, e in self._expiry.items() if e < now]
        for k in dead:
            self.pop(k, None)
            self._expiry.pop(k, None)
