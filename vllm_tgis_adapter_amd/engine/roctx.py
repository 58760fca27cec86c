"""rocTX range markers around engine phases (tracing).

The reference only *propagates* OTel trace headers (its profiler story is
vLLM's); the MI355X-idiomatic addition promised in SURVEY.md §5 is rocTX
ranges around engine steps so `rocprofv3 --marker-trace` (or
`--sys-trace`) shows scheduler/forward/sample/postprocess phases aligned
with the kernel timeline.

Enabled with ``VTA_ROCTX=1``; off by default (zero overhead: the public
helpers are rebound to no-ops at import time when disabled or when
``libroctx64.so`` is unavailable, e.g. on CPU-only CI).
"""

from __future__ import annotations

import ctypes
import os
from contextlib import contextmanager

_lib = None
if os.environ.get("VTA_ROCTX", "0") == "1":
    # rocprofv3 is a rocprofiler-sdk tool: it intercepts the SDK marker
    # library, NOT the legacy roctracer libroctx64 (whose ranges it records
    # as zero regions — verified empirically). Prefer the SDK lib.
    for _name in ("librocprofiler-sdk-roctx.so",
                  "librocprofiler-sdk-roctx.so.1",
                  "libroctx64.so", "libroctx64.so.4"):
        try:
            _lib = ctypes.CDLL(_name)
            break
        except OSError:
            continue
    if _lib is not None:
        _lib.roctxRangePushA.argtypes = [ctypes.c_char_p]
        _lib.roctxRangePushA.restype = ctypes.c_int
        _lib.roctxRangePop.argtypes = []
        _lib.roctxRangePop.restype = ctypes.c_int

enabled = _lib is not None


def range_push(name: str) -> None:
    _lib.roctxRangePushA(name.encode())


def range_pop() -> None:
    _lib.roctxRangePop()


@contextmanager
def trace_range(name: str):
    range_push(name)
    try:
        yield
    finally:
        range_pop()


if not enabled:  # rebind to no-ops so the hot path pays nothing
    def range_push(name: str) -> None:  # noqa: F811
        pass

    def range_pop() -> None:  # noqa: F811
        pass

    @contextmanager
    def trace_range(name: str):  # noqa: F811
        yield
