"""Timing helpers (reference parity: the ``timing`` decorator,
``/root/reference/llama3.2_model.py:12-26`` — defined there but never
enabled; here it is usable and GPU-aware)."""

from __future__ import annotations

import functools
import time
from contextlib import contextmanager


def timing(f):
    """Wall-clock print per call (CPU)."""

    @functools.wraps(f)
    def wrap(*args, **kw):
        t0 = time.perf_counter()
        out = f(*args, **kw)
        print(f"func:{f.__name__} took: {time.perf_counter() - t0:.4f} sec")
        return out

    return wrap


@contextmanager
def gpu_timer(label: str = "", sync: bool = True):
    """Context manager timing a GPU region (synchronizes around it)."""
    import torch

    if sync and torch.cuda.is_available():
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    yield
    if sync and torch.cuda.is_available():
        torch.cuda.synchronize()
    print(f"{label or 'region'}: {(time.perf_counter() - t0) * 1e3:.3f} ms")
