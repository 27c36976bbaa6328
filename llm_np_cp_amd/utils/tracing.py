"""Optional roctx range markers (SURVEY §5 tracing/profiling).

When librocprofiler-sdk-roctx (or legacy libroctx64) is present, wrap
engine phases in named ranges so `rocprofv3 --marker-trace` shows
prefill/decode/layer spans.  No-ops silently when the library or a
profiler is absent — zero overhead in production (a pair of C calls
only when tracing is active).
"""

from __future__ import annotations

import ctypes
from contextlib import contextmanager

_lib = None
_tried = False


def _load():
    global _lib, _tried
    if _tried:
        return _lib
    _tried = True
    for name in ("librocprofiler-sdk-roctx.so", "libroctx64.so",
                 "libroctx64.so.4"):
        try:
            L = ctypes.CDLL(name)
            L.roctxRangePushA.argtypes = [ctypes.c_char_p]
            L.roctxRangePop.argtypes = []
            _lib = L
            break
        except OSError:
            continue
    return _lib


@contextmanager
def trace_range(name: str):
    L = _load()
    if L is None:
        yield
        return
    L.roctxRangePushA(name.encode())
    try:
        yield
    finally:
        L.roctxRangePop()
